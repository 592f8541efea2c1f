from .synthetic import (  # noqa: F401
    SyntheticAN4,
    SyntheticImages,
    SyntheticPTB,
    make_dataset,
    make_loader,
)
from .packed import PackedTensorDataset, pack_dataset  # noqa: F401
