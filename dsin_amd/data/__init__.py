from .provider import Dataset, SyntheticStereo, read_pair_list
from .png import read_png, write_png

__all__ = ["Dataset", "SyntheticStereo", "read_pair_list", "read_png", "write_png"]
