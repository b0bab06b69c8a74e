"""Facade running one optimization pass over a System.

Parity with /root/reference/pkg/manager/manager.go minus the singleton
assignment (``core.TheSystem = system``) — the system is held by the
manager and passed down explicitly.
"""

from __future__ import annotations

from ..core import System
from .optimizer import Optimizer


class Manager:
    def __init__(self, system: System, optimizer: Optimizer) -> None:
        self.system = system
        self.optimizer = optimizer

    def optimize(self) -> None:
        self.optimizer.optimize(self.system)
        self.system.allocate_by_type()
