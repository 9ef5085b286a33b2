from .dataset import Dataset  # noqa: F401
from .synthetic import ShapesDataset  # noqa: F401
from .loader import TorchMaskRCNNDataset, collate_batch, make_data_loader  # noqa: F401
