"""Multi-worker parallelism: one process per GPU, RCCL over xGMI.

Replaces the reference's timely communication crate:
  * data shuffle (the exchange pact, pact.rs:56) → all-to-all-v of delta
    batches by key shard (low 16 bits of the 128-bit key, value.rs:38)
  * progress/frontier gossip → small min-allreduce per step
  * broadcast (external index streams) → dist.broadcast

Backend is torch.distributed: "nccl" IS RCCL on ROCm; CPU tests run the
same code over gloo (gloo lacks all_to_all, so a gather-based fallback is
used there — the NCCL path uses all_to_all_single over xGMI).
"""

from __future__ import annotations

import os
import pickle
from typing import Sequence

import torch
import torch.distributed as dist


class Comm:
    def __init__(self, backend: str | None = None, device=None):
        if not dist.is_initialized():
            backend = backend or ("nccl" if torch.cuda.is_available() else "gloo")
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29571")
            dist.init_process_group(backend=backend)
        self.backend = dist.get_backend()
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        if device is None:
            # honor the engine device (PW_DEVICE) — a CPU-engine worker on
            # a GPU host must not stage exchanges through cuda
            from pathway_amd.internals.config import get_device

            env_dev = get_device()
            if torch.device(env_dev).type == "cuda" and torch.cuda.is_available():
                device = torch.device(
                    f"cuda:{self.rank % max(torch.cuda.device_count(), 1)}"
                )
            else:
                device = torch.device(env_dev)
        self.device = torch.device(device)
        self._comm_device = (
            self.device if str(self.backend) == "nccl" else torch.device("cpu")
        )

    # ---- frontier sync (control plane) ----

    def allreduce_min_time(self, local: int | None) -> int | None:
        BIG = 2**62
        t = torch.tensor(
            [local if local is not None else BIG],
            dtype=torch.int64,
            device=self._comm_device,
        )
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        v = int(t.item())
        return None if v >= BIG else v

    def barrier(self):
        dist.barrier()

    # ---- data plane ----

    def all_to_all_tensor(
        self, tensor: torch.Tensor, send_counts: torch.Tensor
    ) -> torch.Tensor:
        """Rows of `tensor` are grouped by destination rank (sorted);
        send_counts[r] rows go to rank r.  Returns the received rows
        (grouped by source rank)."""
        world = self.world
        send_counts_cpu = send_counts.to("cpu", torch.int64)
        if str(self.backend) == "nccl":
            comm_t = tensor.to(self._comm_device)
            # exchange counts
            recv_counts = torch.empty(world, dtype=torch.int64, device=self._comm_device)
            dist.all_to_all_single(
                recv_counts, send_counts_cpu.to(self._comm_device)
            )
            recv_counts_cpu = recv_counts.cpu()
            out_shape = list(tensor.shape)
            out_shape[0] = int(recv_counts_cpu.sum())
            out = torch.empty(out_shape, dtype=tensor.dtype, device=self._comm_device)
            dist.all_to_all_single(
                out,
                comm_t.contiguous(),
                output_split_sizes=recv_counts_cpu.tolist(),
                input_split_sizes=send_counts_cpu.tolist(),
            )
            return out.to(tensor.device)
        # gloo fallback: gather everything + select own part
        meta = [None] * world
        splits = send_counts_cpu.tolist()
        offs = [0]
        for s in splits:
            offs.append(offs[-1] + s)
        my_parts = [tensor[offs[r] : offs[r + 1]].cpu() for r in range(world)]
        gathered: list[list[torch.Tensor]] = [None] * world  # type: ignore[list-item]
        dist.all_gather_object(gathered, my_parts)
        mine = [gathered[src][self.rank] for src in range(world)]
        out = torch.cat(mine) if mine else tensor[:0].cpu()
        return out.to(tensor.device)

    def all_to_all_objects(self, parts: list) -> list:
        """parts[r] = python object for rank r; returns received objects."""
        world = self.world
        gathered: list[list] = [None] * world  # type: ignore[list-item]
        dist.all_gather_object(gathered, parts)
        return [gathered[src][self.rank] for src in range(world)]

    def allreduce_sum_scalar(self, v: float) -> float:
        t = torch.tensor([v], dtype=torch.float64, device=self._comm_device)
        dist.all_reduce(t)
        return float(t.item())

    def allreduce_max_scalar(self, v: float) -> float:
        """Global max (watermark sync for time-column operators)."""
        t = torch.tensor([v], dtype=torch.float64, device=self._comm_device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())


_COMM: Comm | None = None


def get_comm() -> Comm | None:
    return _COMM


def init(backend: str | None = None, device=None) -> Comm:
    """Initialize multi-worker mode; wires the comm into the run graph."""
    global _COMM
    if _COMM is None:
        _COMM = Comm(backend, device)
        from pathway_amd.internals.config import pathway_config
        from pathway_amd.internals.license import check_worker_limit
        from pathway_amd.internals.rungraph import G

        check_worker_limit(_COMM.world, pathway_config.license_key)
        G.comm = _COMM
    return _COMM


def world_size() -> int:
    c = get_comm()
    return c.world if c else 1
