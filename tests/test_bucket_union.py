"""BucketUnion unit tests (reference: BucketUnionTest.scala /
index/execution/BucketUnionExec.scala — partition-aligned union
preserving HashPartitioning).

The union must: keep bucket alignment (row's murmur3 bucket == its
segment), keep each merged bucket sorted by the indexed column,
preserve the global row multiset, and hash-shuffle (exactly one
recorded shuffle) any unbucketed child on the fly."""

import numpy as np
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd import ops
from hyperspace_amd.execution.columnar import ColumnBatch
from hyperspace_amd.execution.executor import Executor
from hyperspace_amd.plan.nodes import BucketUnionNode, LogicalPlan


class _Leaf(LogicalPlan):
    """Test-only leaf carrying a prepared (batch, seg) pair."""

    def __init__(self, batch, seg):
        super().__init__([])
        self.payload = (batch, seg)

    def label(self):
        return "leaf"


def _bucketed(rng, n, nb):
    from hyperspace_amd.index.covering.index import \
        sort_by_bucket_and_keys
    batch = ColumnBatch({
        "key": torch.from_numpy(rng.integers(0, 5000, n)),
        "val": torch.from_numpy(rng.random(n)),
    })
    bids = ops.murmur3_bucket([batch.tensor("key")], nb)
    return sort_by_bucket_and_keys(batch, bids, ["key"], nb)


@pytest.mark.parametrize("nb", [8, 32])
def test_bucket_union_alignment_and_content(nb):
    rng = np.random.default_rng(61)
    session = hs.HyperspaceSession(device="cpu")
    ex = Executor(session)
    b1, s1 = _bucketed(rng, 40_000, nb)
    # child 2 arrives UNBUCKETED (hybrid-scan appended data)
    b2 = ColumnBatch({
        "key": torch.from_numpy(rng.integers(0, 5000, 7_000)),
        "val": torch.from_numpy(rng.random(7_000)),
    })
    node = BucketUnionNode([_Leaf(b1, s1), _Leaf(b2, None)],
                           nb, ["key"])
    orig_exec = Executor._exec

    def fake_exec(self, plan):
        if isinstance(plan, _Leaf):
            return plan.payload
        return orig_exec(self, plan)

    Executor._exec = fake_exec
    try:
        out, seg = ex._exec_bucket_union(node)
    finally:
        Executor._exec = orig_exec
    assert ex.stats.shuffles == 1  # exactly the unbucketed child
    assert seg.numel() == nb + 1
    assert out.num_rows == 47_000
    keys = out.tensor("key")
    # multiset preserved
    assert int(keys.sum()) == int(b1.tensor("key").sum()
                                  + b2.tensor("key").sum())
    # alignment + within-bucket sortedness
    bids = ops.murmur3_bucket([keys], nb)
    for b in range(nb):
        lo, hi = int(seg[b]), int(seg[b + 1])
        if hi > lo:
            assert (bids[lo:hi] == b).all(), b
            kk = keys[lo:hi]
            assert (kk[1:] >= kk[:-1]).all(), b


def test_bucket_union_empty_child():
    rng = np.random.default_rng(3)
    session = hs.HyperspaceSession(device="cpu")
    ex = Executor(session)
    nb = 8
    b1, s1 = _bucketed(rng, 10_000, nb)
    empty = ColumnBatch({"key": torch.empty(0, dtype=torch.int64),
                         "val": torch.empty(0, dtype=torch.float64)})
    node = BucketUnionNode([_Leaf(b1, s1),
                            _Leaf(empty,
                                  torch.zeros(nb + 1,
                                              dtype=torch.int64))],
                           nb, ["key"])
    orig_exec = Executor._exec
    Executor._exec = lambda self, p: (p.payload
                                      if isinstance(p, _Leaf)
                                      else orig_exec(self, p))
    try:
        out, seg = ex._exec_bucket_union(node)
    finally:
        Executor._exec = orig_exec
    assert out.num_rows == 10_000
    assert seg.numel() == nb + 1
