from .dist import init_distributed, get_rank, get_world_size, barrier
from .flat import FlatParamSpace
from .ddp import BucketedDDP

__all__ = ["init_distributed", "get_rank", "get_world_size", "barrier",
           "FlatParamSpace", "BucketedDDP"]
