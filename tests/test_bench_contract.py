"""Driver-contract test: bench.py must emit exactly one parseable JSON line
with the agreed schema (BASELINE.json metric, value semantics, config block).
Runs the CPU backend on a tiny fleet so it finishes in seconds without a GPU.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(*extra):
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--backend", "cpu",
         "--models-per-gpu", "4", "--steps", "2", "--warmup", "1", *extra],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got {len(lines)}"
    return json.loads(lines[0])


class TestBenchContract:
    def test_schema(self):
        d = run_bench()
        # required fields of the driver contract
        for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                    "ms_per_step", "higher_is_better", "scaling",
                    "vs_baseline", "dtype", "data", "config"):
            assert key in d, f"missing {key}"
        assert d["n_gpus"] == 1
        assert d["steps"] == 2 and d["warmup"] == 1
        assert d["higher_is_better"] is True
        assert d["scaling"] == "weak"
        assert d["unit"] == "allocations/sec"
        assert d["value"] > 0
        assert d["ms_per_step"] > 0
        assert "synthetic" in d["data"]
        cfg = d["config"]
        for key in ("model", "global_batch", "parallelism", "reconcile_p50_ms",
                    "cells_per_step", "analyzer", "solver"):
            assert key in cfg, f"missing config.{key}"
        # whole-job aggregate: value = cells_per_step * steps / elapsed
        assert cfg["cells_per_step"] == 4 * 3  # 4 models x 3 accelerators
        assert d["value"] == pytest.approx(
            cfg["cells_per_step"] * d["steps"] / (d["ms_per_step"] * d["steps"] / 1000.0),
            rel=0.01,
        )

    def test_mg1_flag(self):
        d = run_bench("--cpu-analyzer", "mg1")
        assert d["config"]["analyzer"] == "mg1"

    def test_limited_flag(self):
        d = run_bench("--limited", "8")
        assert "greedy limited" in d["config"]["solver"]


class TestBenchDistributedContract:
    """The driver launches bench.py through torch.distributed.run for N>1 —
    validate that exact invocation on CPU (gloo, world 2)."""

    def test_torchrun_two_ranks(self):
        env = dict(os.environ)
        env["INFERNO_DIST_BACKEND"] = "gloo"
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29921",
             os.path.join(REPO, "bench.py"), "--gpus", "2", "--backend", "cpu",
             "--models-per-gpu", "4", "--steps", "2", "--warmup", "1"],
            capture_output=True, text=True, timeout=600, cwd=REPO, env=env,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
        assert len(lines) == 1, f"expected one JSON line (rank 0 only): {lines}"
        d = json.loads(lines[0])
        assert d["n_gpus"] == 2
        # weak scaling: 2 ranks x 4 models x 3 accelerators
        assert d["config"]["cells_per_step"] == 2 * 4 * 3
        assert "dp2" in d["config"]["parallelism"]

    def test_torchrun_eight_ranks(self):
        """World-8 contract test of the driver's exact launch (VERDICT r1
        item 3): 8 gloo ranks on CPU, tiny fleet, one JSON line from rank 0
        with the whole-job aggregate."""
        env = dict(os.environ)
        env["INFERNO_DIST_BACKEND"] = "gloo"
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
             "--master-port", "29925",
             os.path.join(REPO, "bench.py"), "--gpus", "8", "--backend", "cpu",
             "--models-per-gpu", "2", "--steps", "2", "--warmup", "1"],
            capture_output=True, text=True, timeout=900, cwd=REPO, env=env,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
        assert len(lines) == 1, f"expected one JSON line (rank 0 only): {lines}"
        d = json.loads(lines[0])
        assert d["n_gpus"] == 8
        assert d["config"]["cells_per_step"] == 8 * 2 * 3
        assert "dp8" in d["config"]["parallelism"]


class TestLocalRankMapping:
    """LOCAL_RANK -> HIP device mapping used by bench.py under torchrun."""

    def test_mapping_wraps_device_count(self):
        # 8 local ranks on a 1-GPU box all map to device 0; on an 8-GPU node
        # each rank gets its own device
        for n_dev, expect in ((1, [0] * 8), (4, [0, 1, 2, 3, 0, 1, 2, 3]),
                              (8, list(range(8)))):
            got = [lr % n_dev for lr in range(8)]
            assert got == expect


class TestStrongScaling:
    def test_strong_mode_fixed_total_fleet(self):
        """--scaling strong: the TOTAL fleet stays models-per-gpu as world
        grows (the driver computes the strong-scaling curve from this)."""
        env = dict(os.environ)
        env["INFERNO_DIST_BACKEND"] = "gloo"
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
             "--master-port", "29928",
             os.path.join(REPO, "bench.py"), "--gpus", "2", "--backend", "cpu",
             "--scaling", "strong", "--models-per-gpu", "8",
             "--steps", "2", "--warmup", "1"],
            capture_output=True, text=True, timeout=600, cwd=REPO, env=env,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        d = json.loads([l for l in proc.stdout.splitlines() if l.startswith("{")][0])
        assert d["scaling"] == "strong"
        # total cells = 8 models x 3 accs regardless of world size
        assert d["config"]["cells_per_step"] == 8 * 3
