#!/usr/bin/env python3
"""Flagship benchmark: optimizer allocations/sec + reconcile p50 latency on a
512-model-per-GPU synthetic fleet (BASELINE.json north-star metric).

A "step" is one full reconcile of the autoscaler's compute path: update
per-server loads from a bursty Poisson trace, build the SoA cell snapshot,
run the HIP allocate-sweep kernel (one workgroup per (server, accelerator)
cell: state-dependent M/M/1/K chain + dual SLO bisections + replica/cost
sizing), the segmented-argmin solver kernel, all-gather the per-server
winners across ranks (RCCL over xGMI for N>1), and apply the solution.

Weak scaling: each GPU owns a 512-model shard (N GPUs -> 512*N-model fleet,
matching BASELINE config 5's "4096 models sharded across 8 MI355X").

Single line of JSON on rank 0 per the driver contract.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

# one HW queue per sweep bucket stream so the overlapped bucket launches
# don't serialize (ROCm defaults to 4 queues; must be set before HIP init)
os.environ.setdefault("GPU_MAX_HW_QUEUES", "8")


def _parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--models-per-gpu", type=int, default=512)
    p.add_argument("--backend", choices=["auto", "gpu", "cpu"], default="auto")
    p.add_argument("--seed", type=int, default=1234)
    p.add_argument(
        "--cpu-analyzer",
        dest="analyzer",
        choices=["mm1k", "mg1"],
        default=None,
        help="queueing evaluator: mm1k = reference state-dependent chain "
        "(default), mg1 = closed-form M/G/1/K (BASELINE config 4)",
    )
    p.add_argument(
        "--limited",
        type=int,
        default=0,
        metavar="CAP",
        help="limited mode: greedy capacity-constrained solver with CAP units "
        "per accelerator type (0 = unlimited argmin, the default)",
    )
    p.add_argument(
        "--scaling",
        choices=["weak", "strong"],
        default="weak",
        help="weak (default): each GPU owns a models-per-gpu shard, fleet "
        "grows with N. strong: the TOTAL fleet is fixed at models-per-gpu "
        "and split across ranks",
    )
    p.add_argument(
        "--preset",
        choices=["config2", "config3", "config4", "config5"],
        default=None,
        help="BASELINE.json sweep configs: config2 = 8-model MI355X-only; "
        "config3 = 64 models x 4 TP variants; config4 = 512-model fleet "
        "across MI300X/MI325X/MI355X (default shape); config5 = 4096 models "
        "x 8 variants (sharded)",
    )
    return p.parse_args()


def preset_fleet(preset, models_per_gpu, world, seed):
    """Build the fleet spec for a BASELINE config preset."""
    from inferno_amd.perfmodel import MI355X, accelerator_spec
    from inferno_amd.utils.synthetic import AMD_ACCELERATORS, make_fleet_spec

    if preset == "config2":
        return make_fleet_spec(8 * world, seed=seed,
                               accelerators=[AMD_ACCELERATORS[2]]), 8
    if preset == "config3":
        accs = [accelerator_spec(MI355X, tp) for tp in (1, 2, 4, 8)][:4]
        return make_fleet_spec(64 * world, seed=seed, accelerators=accs), 64
    if preset == "config4":
        spec = make_fleet_spec(512 * world, seed=seed)
        spec.optimizer.analyzer = "mg1"  # config 4 names the M/G/1 model
        return spec, 512
    if preset == "config5":
        # 8 variants per model: the 3-SKU ladder + MI355X TP 2/4/8 + 2 spot tiers
        from dataclasses import replace

        accs = (
            list(AMD_ACCELERATORS)
            + [accelerator_spec(MI355X, tp) for tp in (2, 4, 8)]
            + [
                replace(AMD_ACCELERATORS[0], name="MI300X-spot", cost=45.0),
                replace(AMD_ACCELERATORS[2], name="MI355X-spot", cost=70.0),
            ]
        )
        return make_fleet_spec(4096 * world, seed=seed, accelerators=accs), 4096
    return make_fleet_spec(models_per_gpu * world, seed=seed), models_per_gpu


def main():
    args = _parse_args()
    import torch
    import torch.distributed as dist

    from inferno_amd.config import ServerLoadSpec
    from inferno_amd.core import allocation_from_data
    from inferno_amd.core.system import System
    from inferno_amd.engine import SweepEngine
    from inferno_amd.parallel import ShardedSolver, shard_servers
    from inferno_amd.utils.synthetic import PoissonTrace, make_fleet_spec

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    use_dist = world > 1
    use_gpu = args.backend != "cpu" and torch.cuda.is_available()
    if args.backend == "gpu" and not torch.cuda.is_available():
        print("ERROR: --backend gpu requested but no GPU available", file=sys.stderr)
        sys.exit(2)

    if use_dist:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29771")
        # INFERNO_DIST_BACKEND=gloo lets multi-rank GPU runs share one device
        # (testing); production multi-GPU uses RCCL ("nccl" on ROCm)
        backend = os.environ.get("INFERNO_DIST_BACKEND") or ("nccl" if use_gpu else "gloo")
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
        if use_gpu:
            n_dev = max(torch.cuda.device_count(), 1)
            local_rank = int(os.environ.get("LOCAL_RANK", rank))
            torch.cuda.set_device(local_rank % n_dev)

    fleet_world = 1 if args.scaling == "strong" else world
    spec, args.models_per_gpu = preset_fleet(
        args.preset, args.models_per_gpu, fleet_world, args.seed
    )
    if args.analyzer:
        spec.optimizer.analyzer = args.analyzer
    if args.limited > 0:
        from inferno_amd.config import AcceleratorCount, OptimizerSpec

        spec.optimizer = OptimizerSpec(
            unlimited=False,
            saturationPolicy="PriorityRoundRobin",
            analyzer=spec.optimizer.analyzer,
            analyzerCV2=spec.optimizer.analyzerCV2,
        )
        spec.capacity = [
            AcceleratorCount(type=a.type, count=args.limited) for a in spec.accelerators
        ]
    system, opt_spec = System.from_spec(spec)
    for acc in system.accelerators.values():
        acc.calculate()

    engine = SweepEngine(backend="gpu" if use_gpu else "cpu")
    if use_gpu:
        from inferno_amd.ops.sweep import load_library

        load_library(allow_build=False)  # fail loudly if the in-tree .so is missing
    solver = ShardedSolver(engine)

    import numpy as np

    all_names = sorted(system.servers)
    local_names = shard_servers(all_names, rank, world)
    local_gidx = np.arange(rank, len(all_names), world)
    trace = PoissonTrace(len(all_names), seed=args.seed + 7)
    # pre-sample the whole trace: arrival sampling is workload GENERATION,
    # not reconcile work (the controller starts from collected loads)
    trace_rates = [trace.rates_at(s) for s in range(args.warmup + args.steps)]
    in_toks = np.array(
        [system.servers[n].load.avgInTokens for n in all_names], dtype=np.int32
    )
    out_toks = np.array(
        [system.servers[n].load.avgOutTokens for n in all_names], dtype=np.int32
    )
    # steady-state current-allocation arrays (penalty inputs), local shard
    acc_index = {n: i for i, n in enumerate(sorted(system.accelerators))}
    cur_acc = np.empty(len(local_names), dtype=np.int32)
    cur_rep = np.zeros(len(local_names), dtype=np.int32)
    cur_cost = np.zeros(len(local_names), dtype=np.float32)
    for j, name in enumerate(local_names):
        cur = system.servers[name].cur_allocation
        cur_acc[j] = -3 if cur is None else acc_index.get(cur.accelerator, -1)
        if cur is not None:
            cur_rep[j] = cur.num_replicas
            cur_cost[j] = cur.cost
    state = {"cur": (cur_acc, cur_rep, cur_cost)}

    def reconcile(step: int):
        rates = trace_rates[step]
        fs = solver.fast_sweep
        if use_gpu and fs is not None:
            # steady state: loads + currents flow as arrays (no object churn)
            fs.load_override = (
                rates[local_gidx].astype(np.float32),
                in_toks[local_gidx],
                out_toks[local_gidx],
            )
            fs.cur_override = state["cur"]
        else:
            for i, name in enumerate(all_names):
                system.servers[name].load = ServerLoadSpec(
                    arrivalRate=float(rates[i]),
                    avgInTokens=int(in_toks[i]),
                    avgOutTokens=int(out_toks[i]),
                )
        result = solver.solve(system, opt_spec)
        # apply: desired -> current (HPA/actuation convergence between ticks)
        w = result.winners
        acc = w.acc_idx[local_gidx]
        ok = w.valid[local_gidx] & (acc != -1)
        pa, pr, pc = state["cur"]
        state["cur"] = (
            np.where(ok, np.where(acc == -2, -1, acc), pa).astype(np.int32),
            np.where(ok, w.num_replicas[local_gidx], pr).astype(np.int32),
            np.where(ok, w.cost[local_gidx], pc).astype(np.float32),
        )
        if not (use_gpu and fs is not None):
            for name, data in result.solution.items():
                server = system.servers[name]
                server.spec.currentAlloc = data
                server.cur_allocation = allocation_from_data(data)
        return result

    cells_per_step = None
    # warmup
    for w in range(args.warmup):
        r = reconcile(w)
        cells_per_step = r.local_stats.n_cells

    def barrier_sync():
        if use_dist:
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    barrier_sync()
    step_times = []
    t_start = time.perf_counter()
    for k in range(args.steps):
        t0 = time.perf_counter()
        r = reconcile(args.warmup + k)
        if use_gpu:
            torch.cuda.synchronize()
        step_times.append((time.perf_counter() - t0) * 1000.0)
        cells_per_step = r.local_stats.n_cells
    barrier_sync()
    elapsed = time.perf_counter() - t_start

    # max over ranks (the contract) for elapsed and per-step latencies
    if use_dist:
        dev = "cuda" if (use_gpu and dist.get_backend() == "nccl") else "cpu"
        t = torch.tensor([elapsed] + step_times, dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0].item())
        step_times = [float(x) for x in t[1:].tolist()]
        cells_t = torch.tensor([float(cells_per_step)], dtype=torch.float64, device=dev)
        dist.all_reduce(cells_t, op=dist.ReduceOp.SUM)
        total_cells_per_step = int(cells_t.item())
    else:
        total_cells_per_step = cells_per_step

    allocations_per_sec = total_cells_per_step * args.steps / elapsed
    p50 = statistics.median(step_times)
    p95 = sorted(step_times)[max(int(0.95 * len(step_times)) - 1, 0)]
    ms_per_step = elapsed / args.steps * 1000.0

    result = {
        "metric": "optimizer allocations/sec, 512-model fleet synthetic trace",
        "value": round(allocations_per_sec, 2),
        "unit": "allocations/sec",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": args.scaling,
        "vs_baseline": None,
        "dtype": "fp64",
        "data": "synthetic (random-init perf profiles, bursty Poisson trace)",
        "config": {
            "model": "512-model-fleet x3 AMD accelerator variants (MI300X/MI325X/MI355X)",
            "models_per_gpu": args.models_per_gpu,
            "global_batch": total_cells_per_step,
            "seq_len": None,
            "parallelism": f"dp{world} server-sharded, RCCL allgather winners",
            "backend": engine.backend,
            "reconcile_p50_ms": round(p50, 3),
            "reconcile_p95_ms": round(p95, 3),
            "cells_per_step": total_cells_per_step,
            "analyzer": spec.optimizer.analyzer,
            "solver": (
                ("greedy limited (PriorityRoundRobin)" if args.limited > 0
                 else "unlimited argmin")
                + (" (HIP wva_sweep + wva_argmin)" if engine.backend == "gpu"
                   else " (CPU golden)")
            ),
        },
    }
    if rank == 0:
        print(json.dumps(result))
    if use_dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
