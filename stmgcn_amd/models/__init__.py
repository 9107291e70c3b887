from .stmgcn import GCN, CGRNNCellParams, CG_LSTM, ST_MGCN, StackedSTMGCN, build_model  # noqa: F401
