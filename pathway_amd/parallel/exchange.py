"""Delta-batch exchange: shard-partition + all-to-all-v of column bundles.

Device-side partition (argsort by destination) + per-tensor
all_to_all_single over RCCL/xGMI; host object columns ride a pickled
all-to-all.  The reference's ExchangeCore (pact.rs:56) analog.
"""

from __future__ import annotations

from typing import Any

import torch

from pathway_amd.engine.column import (
    Column,
    ObjectColumn,
    PointerColumn,
    StringColumn,
    TensorColumn,
)
from pathway_amd.internals.api import SHARD_MASK


def shard_of(keys: torch.Tensor, world: int) -> torch.Tensor:
    """Destination rank per row: low 16 key bits mod world (value.rs:38)."""
    return (keys[:, 0] & SHARD_MASK) % world


def exchange_bundle(
    comm,
    dest: torch.Tensor,
    tensors: dict[str, torch.Tensor],
    columns: dict[str, Column],
) -> tuple[dict[str, torch.Tensor], dict[str, Column]]:
    """Exchange rows by dest; tensors dict = named per-row tensors (first dim
    n); columns = engine Columns.  Returns received (tensors, columns)."""
    world = comm.world
    perm = torch.argsort(dest)
    counts = torch.bincount(dest, minlength=world)
    out_tensors: dict[str, torch.Tensor] = {}
    for name, t in tensors.items():
        out_tensors[name] = comm.all_to_all_tensor(
            t.index_select(0, perm).contiguous(), counts
        )
    out_columns: dict[str, Column] = {}
    for name, c in columns.items():
        out_columns[name] = _exchange_column(comm, c, perm, counts)
    return out_tensors, out_columns


def _exchange_column(comm, col: Column, perm: torch.Tensor, counts: torch.Tensor) -> Column:
    world = comm.world
    if isinstance(col, TensorColumn):
        t = comm.all_to_all_tensor(col.tensor.index_select(0, perm).contiguous(), counts)
        mask = None
        if col.mask is not None:
            mask = comm.all_to_all_tensor(
                col.mask.index_select(0, perm).to(torch.uint8).contiguous(), counts
            ).to(torch.bool)
        return TensorColumn(t, col.dtype, mask)
    if isinstance(col, PointerColumn):
        p = comm.all_to_all_tensor(col.pairs.index_select(0, perm).contiguous(), counts)
        return PointerColumn(p, col.dtype)
    if isinstance(col, StringColumn) and getattr(col.pool, "synchronized", False):
        codes = comm.all_to_all_tensor(
            col.codes.index_select(0, perm).contiguous(), counts
        )
        return StringColumn(codes, col.pool, col.dtype)
    # host path: ship the actual values
    vals = col.take(perm).to_pylist()
    offs = [0]
    for c in counts.cpu().tolist():
        offs.append(offs[-1] + c)
    parts = [vals[offs[r] : offs[r + 1]] for r in range(world)]
    received = comm.all_to_all_objects(parts)
    flat = [v for part in received for v in part]
    if isinstance(col, StringColumn):
        return StringColumn.from_strings(flat, device=col.codes.device)
    from pathway_amd.engine.column import column_from_pylist

    return column_from_pylist(flat, col.dtype, device="cpu")
