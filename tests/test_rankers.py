"""Ranker unit tests with hand-built candidates (reference:
FilterIndexRankerTest.scala, JoinIndexRankerTest.scala).

Contracts:
  * FilterIndexRule rank: if ANY candidate needs Hybrid Scan, pick max
    common source bytes; else pick the smallest index
    (FilterIndexRanker.scala:43-64).
  * JoinIndexRule pair rank: equal bucket counts beat unequal, then
    more buckets, then total common bytes; indexed-column order must
    correspond through the join-key pairing
    (JoinIndexRanker.scala:52-90).
"""

import pytest

from hyperspace_amd.rules.candidate_collector import (
    Candidate, TAG_COMMON_SOURCE_SIZE, TAG_HYBRIDSCAN_REQUIRED)
from hyperspace_amd.rules.filter_reason import ReasonCollector
from hyperspace_amd.rules.hyperspace_rules import (FilterIndexRule,
                                                   JoinIndexRule)


class _StubIndex:
    def __init__(self, indexed, buckets):
        self.indexed_columns = indexed
        self.num_buckets = buckets


class _StubEntry:
    def __init__(self, name, index_size, source_size=1000,
                 indexed=("k",), buckets=200):
        self.name = name
        self._index_size = index_size
        self._source_size = source_size
        self.derivedDataset = _StubIndex(list(indexed), buckets)

    def index_files_size(self):
        return self._index_size

    def source_files_size(self):
        return self._source_size


def _cand(name, index_size=100, common=None, hybrid=False,
          indexed=("k",), buckets=200):
    c = Candidate(_StubEntry(name, index_size, indexed=indexed,
                             buckets=buckets))
    if common is not None:
        c.tags[TAG_COMMON_SOURCE_SIZE] = common
    if hybrid:
        c.tags[TAG_HYBRIDSCAN_REQUIRED] = True
    return c


@pytest.fixture()
def filter_rule():
    return FilterIndexRule(None, ReasonCollector())


@pytest.fixture()
def join_rule():
    return JoinIndexRule(None, ReasonCollector())


def test_filter_rank_smallest_index_without_hybrid(filter_rule):
    a = _cand("big", index_size=500)
    b = _cand("small", index_size=50)
    c = _cand("mid", index_size=100)
    assert filter_rule._rank([a, b, c]).name == "small"


def test_filter_rank_max_common_bytes_with_hybrid(filter_rule):
    # the hybrid candidate wins only through common bytes, not size
    a = _cand("stale-small", index_size=10, common=100, hybrid=True)
    b = _cand("fresh-large", index_size=900, common=800, hybrid=True)
    assert filter_rule._rank([a, b]).name == "fresh-large"
    # ANY hybrid candidate switches the whole ranking mode
    c = _cand("exact", index_size=5, common=50)
    assert filter_rule._rank([a, b, c]).name == "fresh-large"


def test_join_rank_prefers_equal_bucket_counts(join_rule):
    l1 = _cand("l200", buckets=200)
    r1 = _cand("r200", buckets=200)
    l2 = _cand("l400", buckets=400)  # more buckets but unequal pair
    lb, rb = join_rule._rank_pairs([l1, l2], [r1], ["k"], ["k"])
    assert (lb.name, rb.name) == ("l200", "r200")


def test_join_rank_prefers_more_buckets_when_both_equal(join_rule):
    l1, r1 = _cand("l100", buckets=100), _cand("r100", buckets=100)
    l2, r2 = _cand("l300", buckets=300), _cand("r300", buckets=300)
    lb, rb = join_rule._rank_pairs([l1, l2], [r1, r2], ["k"], ["k"])
    assert (lb.name, rb.name) == ("l300", "r300")


def test_join_rank_common_bytes_tiebreak(join_rule):
    l1, r1 = _cand("lA", common=10), _cand("rA", common=10)
    l2, r2 = _cand("lB", common=500), _cand("rB", common=500)
    lb, rb = join_rule._rank_pairs([l1, l2], [r1, r2], ["k"], ["k"])
    assert (lb.name, rb.name) == ("lB", "rB")


def test_join_rank_requires_corresponding_column_order(join_rule):
    # join keys pair a<->x, b<->y; left indexed (a,b) matches right
    # (x,y) but NOT right (y,x)
    l = _cand("l", indexed=("a", "b"))
    r_good = _cand("rg", indexed=("x", "y"))
    r_bad = _cand("rb", indexed=("y", "x"))
    lb, rb = join_rule._rank_pairs([l], [r_bad], ["a", "b"], ["x", "y"])
    assert lb is None and rb is None
    lb, rb = join_rule._rank_pairs([l], [r_good, r_bad],
                                   ["a", "b"], ["x", "y"])
    assert (lb.name, rb.name) == ("l", "rg")


def test_join_rank_rejects_arity_mismatch(join_rule):
    l = _cand("l", indexed=("a", "b"))
    r = _cand("r", indexed=("x",))
    lb, rb = join_rule._rank_pairs([l], [r], ["a", "b"], ["x", "y"])
    assert lb is None and rb is None
