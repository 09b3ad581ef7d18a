"""Capacity-constrained greedy solver ("limited mode").

Mirrors the reference's pkg/solver/greedy.go:35-341: per-server sorted
candidate lists with delta-regret ordering, iterative allocation with
re-insertion, and best-effort fallbacks per saturation policy. The reference
keeps this path off production (the controller forces Unlimited,
internal/utils/utils.go:170-173) but the sweep configs 3-5 exercise limited
fleets, so it is implemented fully here. The sequential re-insertion loop is
deliberately host-side (SURVEY.md section 7 "hard parts"): its input — the
sorted candidate allocations — comes from the GPU sweep; the tail is small.
"""
from __future__ import annotations

import bisect
from dataclasses import dataclass, field

from ..config import SaturationPolicy
from ..core import Allocation
from ..core.system import System

_MAX_FLOAT32 = 3.4028234663852886e38


@dataclass
class ServerEntry:
    """Ref: greedy.go:17-24."""

    server_name: str
    priority: int
    cur_index: int = 0
    allocations: list[Allocation] = field(default_factory=list)
    delta: float = 0.0


def _order_key(e: ServerEntry):
    """Sort key equivalent to the reference's orderFunc (greedy.go:76-87):

    priority ascending, then delta descending, then current value descending.
    """
    return (e.priority, -e.delta, -e.allocations[e.cur_index].value)


def solve_greedy(
    system: System,
    delayed_best_effort: bool = False,
    saturation_policy: SaturationPolicy = SaturationPolicy.NONE,
) -> None:
    """Ref: greedy.go:35-104 SolveGreedy."""
    available = dict(system.capacity)

    entries: list[ServerEntry] = []
    for server_name in sorted(system.servers):
        server = system.servers[server_name]
        server.remove_allocation()
        if not server.all_allocations:
            continue
        allocs = sorted(server.all_allocations.values(), key=lambda a: a.value)
        e = ServerEntry(
            server_name=server_name,
            priority=server.priority(system),
            cur_index=0,
            allocations=allocs,
        )
        if len(allocs) > 1:
            e.delta = allocs[1].value - allocs[0].value
        else:
            e.delta = _MAX_FLOAT32
        entries.append(e)

    entries.sort(key=_order_key)

    if delayed_best_effort:
        unallocated = _allocate(system, entries, available)
        _best_effort(system, unallocated, available, saturation_policy)
    else:
        for group in make_priority_groups(entries):
            unallocated = _allocate(system, group, available)
            _best_effort(system, unallocated, available, saturation_policy)


def _allocate(
    system: System, entries: list[ServerEntry], available: dict[str, int]
) -> list[ServerEntry]:
    """Greedy allocation satisfying SLOs; returns unallocated servers.

    Ref: greedy.go:107-166.
    """
    entries = list(entries)
    # parallel sorted key list so re-insertion is a bisect over cached keys
    # instead of re-materializing every key on each pass (the reference's
    # slices.BinarySearchFunc recomputes orderFunc per comparison, greedy.go
    # :158-162 — same ordering, cheaper bookkeeping)
    keys = [_order_key(e) for e in entries]
    unallocated: list[ServerEntry] = []
    while entries:
        top = entries.pop(0)
        keys.pop(0)
        if not top.allocations:
            continue
        server = system.servers.get(top.server_name)
        if server is None:
            continue
        model = system.models.get(server.model_name)
        if model is None:
            continue
        alloc = top.allocations[top.cur_index]
        acc = system.accelerators.get(alloc.accelerator)
        if acc is None:
            continue
        t_name = acc.type
        units_per_replica = model.get_num_instances(acc.name) * acc.multiplicity
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
                top.delta = _MAX_FLOAT32
            k = _order_key(top)
            i = bisect.bisect_left(keys, k)
            entries.insert(i, top)
            keys.insert(i, k)
    return unallocated


def _best_effort(
    system: System,
    unallocated: list[ServerEntry],
    available: dict[str, int],
    policy: SaturationPolicy,
) -> None:
    """Ref: greedy.go:169-191."""
    if policy == SaturationPolicy.PRIORITY_EXHAUSTIVE:
        allocate_maximally(system, unallocated, available)
    elif policy == SaturationPolicy.PRIORITY_ROUND_ROBIN:
        for group in make_priority_groups(unallocated):
            allocate_equally(system, group, available)
    elif policy == SaturationPolicy.ROUND_ROBIN:
        allocate_equally(system, unallocated, available)
    # SaturationPolicy.NONE: nothing beyond satisfying SLOs


def allocate_maximally(
    system: System, server_entries: list[ServerEntry], available: dict[str, int]
) -> None:
    """Exhaustive best-effort in priority order. Ref: greedy.go:194-223."""
    for entry in server_entries:
        server = system.servers.get(entry.server_name)
        if server is None:
            continue
        model = system.models.get(server.model_name)
        if model is None:
            continue
        for alloc in entry.allocations:
            acc = system.accelerators.get(alloc.accelerator)
            if acc is None:
                continue
            units_per_replica = model.get_num_instances(acc.name) * acc.multiplicity
            if units_per_replica <= 0:
                continue
            max_replicas = available.get(acc.type, 0) // units_per_replica
            max_replicas = min(max_replicas, alloc.num_replicas)
            if max_replicas > 0:
                cur = alloc.num_replicas
                factor = float(max_replicas) / float(cur)
                alloc.cost *= factor
                alloc.value *= factor
                alloc.num_replicas = max_replicas
                server.set_allocation(alloc)
                available[acc.type] = available.get(acc.type, 0) - max_replicas * units_per_replica
                break


@dataclass
class _Ticket:
    """Ref: greedy.go:226-236."""

    entry: ServerEntry
    server: object
    model: object
    active: bool = False
    acc_type: str = ""
    units_per_replica: int = 0
    num_replicas: int = 0
    final_alloc: Allocation | None = None


def allocate_equally(
    system: System, server_entries: list[ServerEntry], available: dict[str, int]
) -> None:
    """Round-robin best-effort within a group. Ref: greedy.go:239-316."""
    tickets: dict[str, _Ticket] = {}
    for entry in server_entries:
        server = system.servers.get(entry.server_name)
        model = system.models.get(server.model_name) if server is not None else None
        if server is None or model is None:
            continue
        tickets[entry.server_name] = _Ticket(entry=entry, server=server, model=model)

    allocated: dict[str, _Ticket] = {}
    while tickets:
        for entry in server_entries:
            ticket = tickets.get(entry.server_name)
            if ticket is None:
                continue
            if not ticket.active:
                for alloc in entry.allocations:
                    acc = system.accelerators.get(alloc.accelerator)
                    if acc is None:
                        continue
                    units = ticket.model.get_num_instances(acc.name) * acc.multiplicity
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
                available[ticket.acc_type] = (
                    available.get(ticket.acc_type, 0) - ticket.units_per_replica
                )
                allocated[entry.server_name] = ticket
            else:
                del tickets[entry.server_name]

    for ticket in allocated.values():
        alloc = ticket.final_alloc
        cur = alloc.num_replicas
        factor = float(ticket.num_replicas) / float(cur)
        alloc.cost *= factor
        alloc.value *= factor
        alloc.num_replicas = ticket.num_replicas
        ticket.server.set_allocation(alloc)


def make_priority_groups(entries: list[ServerEntry]) -> list[list[ServerEntry]]:
    """Partition ordered entries into same-priority groups. Ref: greedy.go:321-341."""
    groups: list[list[ServerEntry]] = []
    i = 0
    n = len(entries)
    while i < n:
        group = [entries[i]]
        prio = entries[i].priority
        i += 1
        while i < n and entries[i].priority == prio:
            group.append(entries[i])
            i += 1
        groups.append(group)
    return groups
