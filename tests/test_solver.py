"""Solver tests: unlimited argmin paths, greedy limited mode with all
saturation policies, priority groups, round-robin tickets (mirrors the
reference's solver_test.go / greedy_test.go coverage)."""
import pytest

from inferno_amd.config import OptimizerSpec, SaturationPolicy
from inferno_amd.core import Allocation, System
from inferno_amd.solver import (
    Manager,
    Optimizer,
    ServerEntry,
    Solver,
    make_priority_groups,
    solve_greedy,
)
from tests.fixtures import make_spec


def build(**kw):
    spec = make_spec(**kw)
    system, opt = System.from_spec(spec)
    system.calculate()
    return system, opt


class TestUnlimited:
    def test_argmin_by_value(self):
        system, opt = build(n_servers=6, seed=21)
        solver = Solver(opt)
        solver.solve(system)
        for srv in system.servers.values():
            assert srv.allocation is not None
            best = min(srv.all_allocations.values(), key=lambda a: a.value)
            assert srv.allocation.value == pytest.approx(best.value)

    def test_diff_allocation_populated(self):
        system, opt = build(n_servers=3, seed=22)
        solver = Solver(opt)
        solver.solve(system)
        assert set(solver.diff_allocation) == set(system.servers)
        for name, diff in solver.diff_allocation.items():
            srv = system.servers[name]
            assert diff.old_accelerator == srv.cur_allocation.accelerator
            assert diff.new_accelerator == srv.allocation.accelerator
            assert diff.cost_diff == pytest.approx(
                srv.allocation.cost - srv.cur_allocation.cost, rel=1e-6
            )

    def test_no_candidates_no_allocation(self):
        system, opt = build(n_servers=1, seed=23)
        system.servers["srv-0:ns"].all_allocations = {}
        solver = Solver(opt)
        solver.solve(system)
        assert system.servers["srv-0:ns"].allocation is None

    def test_manager_timing(self):
        system, opt = build(n_servers=2, seed=24)
        optimizer = Optimizer(opt)
        Manager(system, optimizer).optimize()
        assert optimizer.solution_time_msec >= 0.0
        assert system.allocation_by_type  # aggregation ran


class TestGreedy:
    def test_unconstrained_capacity_matches_order(self):
        # with huge capacity every server gets its best (min value) allocation
        system, opt = build(
            n_servers=4,
            seed=31,
            unlimited=False,
            capacity={
                "AMD-MI355X-288GB": 100000,
                "AMD-MI325X-256GB": 100000,
                "AMD-MI300X-192GB": 100000,
            },
        )
        solve_greedy(system)
        for srv in system.servers.values():
            best = min(srv.all_allocations.values(), key=lambda a: a.value)
            assert srv.allocation is not None
            assert srv.allocation.value == pytest.approx(best.value)

    def test_zero_capacity_none_policy(self):
        system, opt = build(n_servers=3, seed=32, unlimited=False, capacity={})
        solve_greedy(system, saturation_policy=SaturationPolicy.NONE)
        for srv in system.servers.values():
            assert srv.allocation is None

    def test_capacity_constraint_prefers_high_priority(self):
        # capacity enough for only part of the fleet: priority 1 servers first
        system, opt = build(
            n_servers=6,
            seed=33,
            unlimited=False,
            capacity={
                "AMD-MI355X-288GB": 6,
                "AMD-MI325X-256GB": 6,
                "AMD-MI300X-192GB": 6,
            },
        )
        solve_greedy(system)
        prio_alloc = {1: 0, 10: 0}
        for srv in system.servers.values():
            if srv.allocation is not None:
                prio_alloc[srv.priority(system)] += 1
        assert prio_alloc[1] >= prio_alloc[10]

    def test_capacity_accounting_never_negative(self):
        cap = {"AMD-MI355X-288GB": 9, "AMD-MI325X-256GB": 5, "AMD-MI300X-192GB": 3}
        system, opt = build(n_servers=8, seed=34, unlimited=False, capacity=dict(cap))
        solve_greedy(system)
        used: dict[str, int] = {t: 0 for t in cap}
        for srv in system.servers.values():
            alloc = srv.allocation
            if alloc is None:
                continue
            acc = system.accelerators[alloc.accelerator]
            model = system.models[srv.model_name]
            used[acc.type] += (
                alloc.num_replicas * model.get_num_instances(acc.name) * acc.multiplicity
            )
        for t in cap:
            assert used[t] <= cap[t]

    @pytest.mark.parametrize(
        "policy",
        [
            SaturationPolicy.PRIORITY_EXHAUSTIVE,
            SaturationPolicy.PRIORITY_ROUND_ROBIN,
            SaturationPolicy.ROUND_ROBIN,
        ],
    )
    def test_best_effort_policies_allocate_partial(self, policy):
        # tiny capacity: SLO-satisfying allocation impossible, best effort kicks in
        cap = {"AMD-MI355X-288GB": 2, "AMD-MI325X-256GB": 2, "AMD-MI300X-192GB": 2}
        system, opt = build(n_servers=6, seed=35, unlimited=False, capacity=dict(cap))
        solve_greedy(system, saturation_policy=policy)
        used: dict[str, int] = {t: 0 for t in cap}
        got_any = False
        for srv in system.servers.values():
            alloc = srv.allocation
            if alloc is None:
                continue
            got_any = True
            acc = system.accelerators[alloc.accelerator]
            model = system.models[srv.model_name]
            used[acc.type] += (
                alloc.num_replicas * model.get_num_instances(acc.name) * acc.multiplicity
            )
        assert got_any
        for t in cap:
            assert used[t] <= cap[t]

    def test_best_effort_scales_cost(self):
        # a single server demanding more than capacity gets scaled down with cost factor
        system, opt = build(
            n_servers=1,
            seed=36,
            unlimited=False,
            arrival_scale=6000.0,  # very high load -> many replicas
            capacity={"AMD-MI355X-288GB": 1, "AMD-MI325X-256GB": 1, "AMD-MI300X-192GB": 1},
        )
        srv = system.servers["srv-0:ns"]
        full = {k: a.clone() for k, a in srv.all_allocations.items()}
        solve_greedy(system, saturation_policy=SaturationPolicy.PRIORITY_EXHAUSTIVE)
        alloc = srv.allocation
        if alloc is not None:
            orig = full[alloc.accelerator]
            assert alloc.num_replicas <= orig.num_replicas
            factor = alloc.num_replicas / orig.num_replicas
            assert alloc.cost == pytest.approx(orig.cost * factor, rel=1e-5)

    def test_delayed_best_effort_runs_once_globally(self):
        cap = {"AMD-MI355X-288GB": 4, "AMD-MI325X-256GB": 4, "AMD-MI300X-192GB": 4}
        system, opt = build(
            n_servers=6, seed=37, unlimited=False, delayed_best_effort=True, capacity=dict(cap)
        )
        solve_greedy(
            system,
            delayed_best_effort=True,
            saturation_policy=SaturationPolicy.ROUND_ROBIN,
        )
        # must not over-allocate
        used: dict[str, int] = {t: 0 for t in cap}
        for srv in system.servers.values():
            alloc = srv.allocation
            if alloc is None:
                continue
            acc = system.accelerators[alloc.accelerator]
            model = system.models[srv.model_name]
            used[acc.type] += (
                alloc.num_replicas * model.get_num_instances(acc.name) * acc.multiplicity
            )
        for t in cap:
            assert used[t] <= cap[t]


class TestPriorityGroups:
    def test_grouping(self):
        def entry(p):
            return ServerEntry(server_name=f"s{p}", priority=p, allocations=[Allocation()])

        entries = [entry(1), entry(1), entry(5), entry(10), entry(10), entry(10)]
        groups = make_priority_groups(entries)
        assert [len(g) for g in groups] == [2, 1, 3]
        assert [g[0].priority for g in groups] == [1, 5, 10]

    def test_empty(self):
        assert make_priority_groups([]) == []
