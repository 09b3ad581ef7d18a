"""Engine reentrancy under concurrency (SURVEY.md section 5 race-detection
plan: the reference's core has deliberate mutable globals that are only safe
because reconciles are serialized; this build's engine must be safe to run
concurrently — no TheSystem singleton, no analyzer eval globals)."""
import concurrent.futures

import numpy as np
import pytest

from inferno_amd.core.system import System
from inferno_amd.engine import SweepEngine
from inferno_amd.parallel import ShardedSolver
from tests.fixtures import make_spec


class TestConcurrentSolves:
    def test_parallel_engine_solves_match_serial(self):
        """Four concurrent solves over distinct systems produce exactly the
        serial results (no cross-talk through hidden shared state)."""
        def solve_one(seed):
            system, opt = System.from_spec(make_spec(n_servers=8, seed=seed))
            ShardedSolver(SweepEngine(backend="cpu")).solve(system, opt)
            return {
                n: (s.allocation.accelerator, s.allocation.num_replicas)
                if s.allocation else None
                for n, s in system.servers.items()
            }

        serial = {seed: solve_one(seed) for seed in (11, 22, 33, 44)}
        with concurrent.futures.ThreadPoolExecutor(max_workers=4) as pool:
            futures = {seed: pool.submit(solve_one, seed) for seed in (11, 22, 33, 44)}
            parallel = {seed: f.result() for seed, f in futures.items()}
        assert parallel == serial

    def test_shared_engine_concurrent_sweeps(self):
        """One SweepEngine instance used from two threads on two systems:
        results identical to sequential use (the engine holds no per-solve
        mutable state)."""
        engine = SweepEngine(backend="cpu")

        def sweep_one(seed):
            system, _ = System.from_spec(make_spec(n_servers=6, seed=seed))
            engine.sweep(system)
            return {
                n: {a: al.num_replicas for a, al in s.all_allocations.items()}
                for n, s in system.servers.items()
            }

        ref = [sweep_one(7), sweep_one(8)]
        with concurrent.futures.ThreadPoolExecutor(max_workers=2) as pool:
            got = list(pool.map(sweep_one, (7, 8)))
        assert got == ref

    def test_concurrent_analyzers_are_independent(self):
        """QueueAnalyzer instances used concurrently (the reference's eval
        closures were package globals, queueanalyzer.go:176-179 — a latent
        race this design removed)."""
        from inferno_amd.analyzer import (
            Configuration, DecodeParms, PrefillParms, QueueAnalyzer,
            RequestSize, ServiceParms, TargetPerf,
        )

        def size_one(alpha):
            cfg = Configuration(64, 640, ServiceParms(
                PrefillParms(5.0, 0.01), DecodeParms(alpha, 0.2)))
            qa = QueueAnalyzer(cfg, RequestSize(128, 64))
            _, metrics, _ = qa.size(TargetPerf(target_ttft=2000.0, target_itl=80.0))
            return metrics.throughput

        alphas = [5.0, 10.0, 20.0, 40.0]
        serial = [size_one(a) for a in alphas]
        with concurrent.futures.ThreadPoolExecutor(max_workers=4) as pool:
            parallel = list(pool.map(size_one, alphas))
        assert parallel == serial
