"""Greedy capacity-constrained solver (limited mode).

Parity with /root/reference/pkg/solver/greedy.go: per-server sorted
allocation lists; entries ordered by (priority, regret-delta to next-best,
current value); greedy grant, else advance index and binary-search reinsert;
leftover servers get best-effort allocation per saturation policy.

Ordering invariants preserved exactly:
- entries sort by priority ascending, then delta *descending* (largest
  regret first), then current-allocation value *descending*
  (greedy.go:72-83);
- on a capacity miss the entry's index advances and its delta becomes the
  gap to the next-best allocation, or +inf for the last choice
  (greedy.go:147-163);
- best-effort policies may scale an allocation's replica count and rescale
  cost/value by the replica factor (greedy.go:208-211,299-310).
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

from ..config import OptimizerSpec, SaturationPolicy
from ..core import Allocation, System


class ServerEntry:
    __slots__ = ("server_name", "priority", "cur_index", "allocations", "delta")

    def __init__(self, server_name: str, priority: int, allocations: List[Allocation]) -> None:
        self.server_name = server_name
        self.priority = priority
        self.cur_index = 0
        self.allocations = allocations
        self.delta = 0.0

    def __repr__(self) -> str:
        return (
            f"sName={self.server_name}, prio={self.priority}, curIndex={self.cur_index}, "
            f"delta={self.delta}, allocations={self.allocations}"
        )


def _order_cmp(a: ServerEntry, b: ServerEntry) -> int:
    """Straight priorities, then regret deltas (descending), then current
    value (descending)."""
    if a.priority != b.priority:
        return -1 if a.priority < b.priority else 1
    if a.delta != b.delta:
        return -1 if b.delta < a.delta else 1
    av = a.allocations[a.cur_index].value
    bv = b.allocations[b.cur_index].value
    if av == bv:
        return 0
    return -1 if bv < av else 1


def _insort(entries: List[ServerEntry], entry: ServerEntry) -> None:
    """Insert keeping the _order_cmp ordering (leftmost position with
    cmp(entries[i], entry) >= 0, mirroring slices.BinarySearchFunc)."""
    lo, hi = 0, len(entries)
    while lo < hi:
        mid = (lo + hi) // 2
        if _order_cmp(entries[mid], entry) < 0:
            lo = mid + 1
        else:
            hi = mid
    entries.insert(lo, entry)


def solve_greedy(system: System, spec: OptimizerSpec) -> None:
    available: Dict[str, int] = dict(system.capacity)

    entries: List[ServerEntry] = []
    for server_name, server in system.servers.items():
        server.remove_allocation()
        all_allocs = server.all_allocations
        if not all_allocs:
            continue
        allocs = sorted(all_allocs.values(), key=lambda a: a.value)
        e = ServerEntry(server_name, server.priority(system), allocs)
        if len(allocs) > 1:
            e.delta = allocs[1].value - allocs[0].value
        else:
            e.delta = math.inf
        entries.append(e)

    # same ordering as _order_cmp — (priority asc, regret delta desc,
    # current value desc) — as a key tuple: ~10x cheaper than the
    # cmp_to_key wrapper, and equally stable for ties
    entries.sort(
        key=lambda e: (e.priority, -e.delta, -e.allocations[e.cur_index].value)
    )

    if spec.delayed_best_effort:
        unallocated = _allocate(system, entries, available)
        _best_effort(system, unallocated, available, spec.saturation_policy)
    else:
        for group in make_priority_groups(entries):
            unallocated = _allocate(system, group, available)
            _best_effort(system, unallocated, available, spec.saturation_policy)


def _allocate(
    system: System, entries: List[ServerEntry], available: Dict[str, int]
) -> List[ServerEntry]:
    """Greedy SLO-satisfying allocation; returns entries that got nothing."""
    entries = list(entries)
    unallocated: List[ServerEntry] = []
    while entries:
        top = entries.pop(0)
        if not top.allocations:
            continue
        server = system.server(top.server_name)
        if server is None:
            continue
        model = system.model(server.model_name)
        if model is None:
            continue
        alloc = top.allocations[top.cur_index]
        acc = system.accelerator(alloc.accelerator)
        if acc is None:
            continue
        t_name = acc.type
        units_per_replica = model.get_num_instances(alloc.accelerator) * acc.multiplicity
        count = alloc.num_replicas * units_per_replica

        if available.get(t_name, 0) >= count:
            available[t_name] = available.get(t_name, 0) - count
            server.set_allocation(alloc)
        else:
            top.cur_index += 1
            if top.cur_index + 1 < len(top.allocations):
                top.delta = (
                    top.allocations[top.cur_index + 1].value
                    - top.allocations[top.cur_index].value
                )
            elif top.cur_index == len(top.allocations):
                unallocated.append(top)
                continue
            else:
                top.delta = math.inf
            _insort(entries, top)
    return unallocated


def _best_effort(
    system: System,
    unallocated: List[ServerEntry],
    available: Dict[str, int],
    policy: str,
) -> None:
    p = SaturationPolicy.parse(policy)
    if p is SaturationPolicy.PRIORITY_EXHAUSTIVE:
        _allocate_maximally(system, unallocated, available)
    elif p is SaturationPolicy.PRIORITY_ROUND_ROBIN:
        for group in make_priority_groups(unallocated):
            _allocate_equally(system, group, available)
    elif p is SaturationPolicy.ROUND_ROBIN:
        _allocate_equally(system, unallocated, available)
    # SaturationPolicy.NONE: no allocation beyond satisfying SLOs


def _allocate_maximally(
    system: System, entries: List[ServerEntry], available: Dict[str, int]
) -> None:
    """Priority ordering: one server at a time exhaustively."""
    for entry in entries:
        for alloc in entry.allocations:
            acc_name = alloc.accelerator
            server = system.server(entry.server_name)
            if server is None:
                continue
            model = system.model(server.model_name)
            acc = system.accelerator(acc_name)
            if acc is None or model is None:
                continue
            units_per_replica = model.get_num_instances(acc_name) * acc.multiplicity
            if units_per_replica <= 0:
                continue
            max_replicas = available.get(acc.type, 0) // units_per_replica
            max_replicas = min(max_replicas, alloc.num_replicas)
            if max_replicas > 0:
                cur = alloc.num_replicas
                factor = max_replicas / cur
                alloc.cost *= factor
                alloc.value *= factor
                alloc.num_replicas = max_replicas
                server.set_allocation(alloc)
                available[acc.type] = available.get(acc.type, 0) - max_replicas * units_per_replica
                break


class _Ticket:
    __slots__ = ("entry", "active", "server", "model", "acc_type", "units_per_replica", "num_replicas", "final_alloc")

    def __init__(self, entry: ServerEntry, server, model) -> None:
        self.entry = entry
        self.active = False
        self.server = server
        self.model = model
        self.acc_type = ""
        self.units_per_replica = 0
        self.num_replicas = 0
        self.final_alloc: Optional[Allocation] = None


def _allocate_equally(
    system: System, entries: List[ServerEntry], available: Dict[str, int]
) -> None:
    """Round-robin: one replica per visit until capacity runs out."""
    tickets: Dict[str, _Ticket] = {}
    for entry in entries:
        server = system.server(entry.server_name)
        if server is None:
            continue
        model = system.model(server.model_name)
        if model is None:
            continue
        tickets[entry.server_name] = _Ticket(entry, server, model)

    allocated: Dict[str, _Ticket] = {}
    while tickets:
        for entry in entries:
            ticket = tickets.get(entry.server_name)
            if ticket is None:
                continue
            if not ticket.active:
                for alloc in entry.allocations:
                    acc = system.accelerator(alloc.accelerator)
                    if acc is None:
                        continue
                    units = ticket.model.get_num_instances(alloc.accelerator) * acc.multiplicity
                    if units > 0 and available.get(acc.type, 0) >= units:
                        ticket.active = True
                        ticket.acc_type = acc.type
                        ticket.units_per_replica = units
                        ticket.final_alloc = alloc
                        break
                if not ticket.active:
                    del tickets[entry.server_name]
                    continue
            replicas_available = available.get(ticket.acc_type, 0) // ticket.units_per_replica
            if min(replicas_available, ticket.final_alloc.num_replicas) > 0:
                ticket.num_replicas += 1
                available[ticket.acc_type] = available.get(ticket.acc_type, 0) - ticket.units_per_replica
                allocated[entry.server_name] = ticket
            else:
                del tickets[entry.server_name]

    for ticket in allocated.values():
        alloc = ticket.final_alloc
        cur = alloc.num_replicas
        factor = ticket.num_replicas / cur
        alloc.cost *= factor
        alloc.value *= factor
        alloc.num_replicas = ticket.num_replicas
        ticket.server.set_allocation(alloc)


def make_priority_groups(entries: List[ServerEntry]) -> List[List[ServerEntry]]:
    """Partition an ordered entry list into runs of equal priority."""
    groups: List[List[ServerEntry]] = []
    i, n = 0, len(entries)
    while i < n:
        group = [entries[i]]
        prio = entries[i].priority
        i += 1
        while i < n and entries[i].priority == prio:
            group.append(entries[i])
            i += 1
        groups.append(group)
    return groups
