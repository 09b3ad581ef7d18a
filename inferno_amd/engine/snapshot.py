"""SoA cell snapshot: System -> flat tensors for the GPU sweep.

One cell per (server, candidate accelerator) pair, grouped contiguously by
server (segment layout consumed by the argmin kernel). Mirrors the
feasibility pre-conditions of the reference's CreateAllocation lookups
(pkg/core/allocation.go:42-70): cells are only emitted where accelerator,
server load, model perf data and service-class target all resolve.
"""
from __future__ import annotations

from dataclasses import dataclass

import numpy as np

from ..core.system import System

FLAG_CUR_SAME = 1
FLAG_CUR_EMPTY = 2
FLAG_HAS_CUR = 4


@dataclass
class CellSnapshot:
    """Flat cell arrays + metadata to map results back to domain objects."""

    arrays: dict  # name -> torch tensor (CPU)
    server_names: list[str]  # per segment
    cell_acc: list[str]  # per cell: accelerator name
    cell_server: np.ndarray  # per cell: segment index (int32)
    seg_start: "object"  # torch int32 [n_servers+1]

    @property
    def n_cells(self) -> int:
        return len(self.cell_acc)


def compute_batch_size(server, perf, K: int) -> int:
    """N from server override or scaled perf data (ref allocation.go:80-86).
    Uncapped like the reference; N > 8192 cells use the GPU's global-memory
    geometry spill path."""
    if server.max_batch_size > 0:
        n = server.max_batch_size
    else:
        n = max(perf.maxBatchSize * perf.atTokens // K, 1)
    return n


def build_cell_snapshot(system: System, server_names: list[str] | None = None) -> CellSnapshot:
    """Build the SoA snapshot for (a shard of) the system's servers.

    ``server_names`` restricts to a shard (multi-GPU partitioning); default
    is all servers in sorted-name order (deterministic cell layout).
    """
    import torch

    if server_names is None:
        server_names = sorted(system.servers)

    ints: dict[str, list[int]] = {
        k: []
        for k in (
            "in_tok",
            "out_tok",
            "batch_n",
            "min_replicas",
            "perf_max_batch",
            "cur_replicas",
            "flags",
        )
    }
    floats: dict[str, list[float]] = {
        k: []
        for k in (
            "alpha",
            "beta",
            "gamma",
            "delta",
            "arrival_rate",
            "t_itl",
            "t_ttft",
            "t_tps",
            "acc_cost",
            "cur_cost",
        )
    }
    cell_acc: list[str] = []
    cell_server: list[int] = []
    seg_start = [0]

    for seg, srv_name in enumerate(server_names):
        server = system.servers[srv_name]
        load = server.load
        model = system.models.get(server.model_name)
        svc = system.service_classes.get(server.service_class_name)
        target = svc.model_target(server.model_name) if svc is not None else None
        valid_server = (
            load is not None
            and load.arrivalRate >= 0
            and load.avgInTokens >= 0
            and load.avgOutTokens >= 0
            and model is not None
            and target is not None
        )
        if valid_server:
            cur = server.cur_allocation
            for acc_name in sorted(server.candidate_accelerators(system)):
                acc = system.accelerators[acc_name]
                perf = model.get_perf_data(acc_name)
                if perf is None:
                    continue
                K = load.avgOutTokens
                zero_load = load.arrivalRate == 0 or K == 0
                n = 1 if zero_load else compute_batch_size(server, perf, K)
                pmb = (
                    server.max_batch_size
                    if server.max_batch_size > 0
                    else perf.maxBatchSize
                )
                flags = 0
                if cur is not None:
                    flags |= FLAG_HAS_CUR
                    if cur.accelerator == acc_name:
                        flags |= FLAG_CUR_SAME
                    if cur.accelerator == "":
                        flags |= FLAG_CUR_EMPTY
                ints["in_tok"].append(load.avgInTokens)
                ints["out_tok"].append(K)
                ints["batch_n"].append(n)
                ints["min_replicas"].append(server.min_num_replicas)
                ints["perf_max_batch"].append(pmb)
                ints["cur_replicas"].append(cur.num_replicas if cur is not None else 0)
                ints["flags"].append(flags)
                floats["alpha"].append(perf.decodeParms.alpha)
                floats["beta"].append(perf.decodeParms.beta)
                floats["gamma"].append(perf.prefillParms.gamma)
                floats["delta"].append(perf.prefillParms.delta)
                floats["arrival_rate"].append(load.arrivalRate)
                floats["t_itl"].append(target.itl)
                floats["t_ttft"].append(target.ttft)
                floats["t_tps"].append(target.tps)
                floats["acc_cost"].append(acc.cost * model.get_num_instances(acc_name))
                floats["cur_cost"].append(cur.cost if cur is not None else 0.0)
                cell_acc.append(acc_name)
                cell_server.append(seg)
        seg_start.append(len(cell_acc))

    arrays = {k: torch.tensor(v, dtype=torch.int32) for k, v in ints.items()}
    arrays.update({k: torch.tensor(v, dtype=torch.float32) for k, v in floats.items()})
    return CellSnapshot(
        arrays=arrays,
        server_names=list(server_names),
        cell_acc=cell_acc,
        cell_server=np.asarray(cell_server, dtype=np.int32),
        seg_start=torch.tensor(seg_start, dtype=torch.int32),
    )
