"""RCCL (nccl-on-ROCm) path exercised on a real MI355X (VERDICT r1 item 3:
the nccl backend path had never run on hardware — every multi-rank test used
gloo). World-1 keeps it single-box-safe; the collective code path (cuda
tensors through dist.all_gather / all_gather_object) is the same one the
driver's 8-GPU run takes."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestWorldOneRccl:
    def test_sharded_solver_over_nccl_group(self):
        """init_process_group('nccl') at world 1 + the full sharded solve:
        first contact for RCCL + per-rank HIP context + cuda-tensor
        collectives."""
        import torch.distributed as dist

        from inferno_amd.core.system import System
        from inferno_amd.engine import SweepEngine
        from inferno_amd.parallel import ShardedSolver
        from tests.fixtures import make_spec

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29951")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        created = False
        if not dist.is_initialized():
            dist.init_process_group(backend="nccl", rank=0, world_size=1)
            created = True
        try:
            torch.cuda.set_device(0)
            sys_gpu, opt = System.from_spec(make_spec(n_servers=16, seed=500))
            sys_cpu, _ = System.from_spec(make_spec(n_servers=16, seed=500))
            solver = ShardedSolver(SweepEngine(backend="gpu"))
            result = solver.solve(sys_gpu, opt)
            ref = ShardedSolver(SweepEngine(backend="cpu")).solve(sys_cpu, opt)
            assert set(result.solution) == set(ref.solution)
            for name in ref.solution:
                a, b = ref.solution[name], result.solution[name]
                assert a.accelerator == b.accelerator, name
                assert abs(a.numReplicas - b.numReplicas) <= 1, name
        finally:
            if created:
                dist.destroy_process_group()

    def test_torchrun_world1_nccl_bench(self):
        """The driver's exact N=1-under-torchrun invocation with the nccl
        backend selected (use_gpu=True path in bench.py:115-125)."""
        env = dict(os.environ)
        env.pop("INFERNO_DIST_BACKEND", None)  # let bench pick nccl on GPU
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
             "--master-port", "29953",
             os.path.join(REPO, "bench.py"), "--gpus", "1",
             "--models-per-gpu", "32", "--steps", "3", "--warmup", "1"],
            capture_output=True, text=True, timeout=600, cwd=REPO, env=env,
        )
        assert proc.returncode == 0, proc.stderr[-2000:]
        lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
        assert len(lines) == 1
        d = json.loads(lines[0])
        assert d["config"]["backend"] == "gpu"
        assert d["value"] > 0
