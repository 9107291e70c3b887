from .preprocess import SupportGenerator, Adj_Preprocessor, CSRSupport  # noqa: F401
