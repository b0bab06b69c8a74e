"""Core domain model (layer L4): System/Accelerator/Model/ServiceClass/Server
and the Allocation sizing kernel.

Parity with /root/reference/pkg/core/, with one deliberate architectural
change: there is **no** ``TheSystem`` package-global singleton
(system.go:10-13).  Every function that needs registry lookups takes the
:class:`System` explicitly, so concurrent optimizations over different
systems are safe by construction.
"""

from .accelerator import Accelerator
from .model import Model
from .serviceclass import ServiceClass, Target
from .allocation import Allocation, AllocationDiff, create_allocation
from .server import Server
from .system import AllocationByType, System

__all__ = [
    "Accelerator",
    "Model",
    "ServiceClass",
    "Target",
    "Allocation",
    "AllocationDiff",
    "create_allocation",
    "Server",
    "AllocationByType",
    "System",
]
