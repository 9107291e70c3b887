from .container import (  # noqa: F401
    DataInput, DataGenerator, TaxiDataset, DeviceLoader,
)
from .synthetic import make_synthetic_dataset, write_synthetic_npz  # noqa: F401
