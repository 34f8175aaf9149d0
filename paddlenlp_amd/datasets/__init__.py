from .dataset import MapDataset, load_dataset  # noqa: F401
from .zero_padding_dataset import ZeroPaddingMapDataset  # noqa: F401
