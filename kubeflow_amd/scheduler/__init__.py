from .inventory import GpuInventory
from .gang import GangScheduler, InsufficientResources
from .launcher import ProcessGang, launch_gang

__all__ = ["GpuInventory", "GangScheduler", "InsufficientResources",
           "ProcessGang", "launch_gang"]
