from .provider import (Dataset, SyntheticStereo, SyntheticDataset,
                       make_dataset, read_pair_list)
from .png import read_png, write_png

__all__ = ["Dataset", "SyntheticStereo", "SyntheticDataset", "make_dataset",
           "read_pair_list", "read_png", "write_png"]
