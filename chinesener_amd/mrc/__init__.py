from .convert import convert2mrc
from .dataset import MrcDataset, build_single_feature

__all__ = ["convert2mrc", "MrcDataset", "build_single_feature"]
