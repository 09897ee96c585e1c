"""Build-time bucket exchange: RCCL all-to-all over xGMI (C1).

Each rank holds a shard of the source rows with computed bucket ids; after
the exchange, rank r holds exactly the rows of its owned buckets
(b % world == r).  xGMI is point-to-point (7 links × ≈153 GB/s per GPU on
an 8-GPU MI355X node), so the pairwise all-to-all moves every shard on its
dedicated link — aggregate ≈7×153 GB/s per GPU — rather than being bound
by a single link like ring collectives.

Implementation: rows are first partitioned locally by destination rank
(one stable sort pass on the device), then exchanged with
``dist.all_to_all_single`` per column (RCCL batches these over xGMI).
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.distributed as dist

from .. import ops
from ..execution.columnar import ColumnBatch, StringColumn


def exchange_by_bucket(batch: ColumnBatch, bucket_ids: torch.Tensor,
                       num_buckets: int
                       ) -> Tuple[ColumnBatch, torch.Tensor]:
    """All-to-all rows to their bucket owners.

    Returns (received batch, received bucket ids).  Deterministic: rows
    arrive grouped by source rank in rank order, preserving source order
    within each rank (RCCL all_to_all concatenates in rank order).
    """
    world = dist.get_world_size()
    rank = dist.get_rank()
    device = batch.device

    dest = (bucket_ids.to(torch.int64) % world)
    # group rows by destination rank: stable sort by dest
    perm = ops.sort_perm(ops.normalize_key(dest))
    dest_sorted = dest[perm]
    batch = batch.gather(perm)
    bucket_sorted = ops.gather_rows(bucket_ids.to(torch.int64), perm)

    send_counts = torch.bincount(dest_sorted.cpu(), minlength=world)
    recv_counts = torch.empty(world, dtype=torch.int64)
    send_list = [int(x) for x in send_counts]
    # exchange counts
    if device.type == "cuda":
        sc = send_counts.to(device)
        rc = torch.empty(world, dtype=torch.int64, device=device)
        dist.all_to_all_single(rc, sc)
        recv_counts = rc.cpu()
    else:
        dist.all_to_all_single(recv_counts, send_counts)
    recv_list = [int(x) for x in recv_counts]
    n_recv = sum(recv_list)

    def a2a(t: torch.Tensor) -> torch.Tensor:
        out = torch.empty(n_recv, dtype=t.dtype, device=t.device)
        dist.all_to_all_single(out, t.contiguous(),
                               output_split_sizes=recv_list,
                               input_split_sizes=send_list)
        return out

    out_cols = {}
    out_masks = {}
    for name, col in batch.columns.items():
        if isinstance(col, StringColumn):
            # dictionary-merge exchange: every rank gathers all local
            # dictionaries (small, host-side pickle collective), remaps
            # its codes onto the merged sorted dictionary, and the i32
            # codes travel like any other column.  The merged dictionary
            # is identical on every rank, so received codes are
            # comparable without further translation.
            local_values = [list(col.values)]
            gathered: List[object] = [None] * world
            dist.all_gather_object(gathered, local_values)
            merged = sorted(set().union(
                *[set(v[0]) for v in gathered]))  # type: ignore[index]
            val_index = {v: i for i, v in enumerate(merged)}
            lut = torch.tensor([val_index[v] for v in col.values],
                               dtype=torch.int32, device=col.codes.device)
            codes = lut[col.codes.long()] if len(col.values) else \
                col.codes
            out_cols[name] = StringColumn(a2a(codes), merged)
        else:
            out_cols[name] = a2a(col)
        m = batch.mask(name)
        if m is not None:
            # validity masks ride along as uint8 splits on the same links
            out_masks[name] = a2a(m.to(torch.uint8)).to(torch.bool)
    out_buckets = a2a(bucket_sorted)
    return ColumnBatch(out_cols, out_masks), out_buckets.to(torch.int32)
