from .synthetic import (  # noqa: F401
    SyntheticAN4,
    SyntheticImages,
    SyntheticPTB,
    make_dataset,
    make_loader,
)
from .packed import PackedTensorDataset, pack_dataset  # noqa: F401
from .audio import (  # noqa: F401
    SpectrogramDataset,
    SyntheticAN4Waveforms,
    labels_to_text,
    spectrogram,
    text_to_labels,
)
