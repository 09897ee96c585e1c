"""Property-based tests (hypothesis) over schemas, configs and
predicates vs numpy/pandas oracles — the randomized counterpart of the
reference's example-based suites (ROADMAP testing item)."""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
from hypothesis import given, settings, strategies as st

import hyperspace_amd as hs
from hyperspace_amd.execution.columnar import ColumnBatch

SETTINGS = dict(max_examples=20, deadline=None)


# ---------------------------------------------------------------------------
# sketch expression grammar: parse/print round-trip + oracle eval
# ---------------------------------------------------------------------------

_cols = st.sampled_from(["a", "b_1", "x.y"])
_lits = st.integers(min_value=-1000, max_value=1000)


def _expr_strings(depth=2):
    if depth == 0:
        return st.one_of(_cols, _lits.map(str))
    sub = _expr_strings(depth - 1)
    return st.one_of(
        _cols, _lits.map(str),
        st.tuples(sub, st.sampled_from("+-*%"), sub).map(
            lambda t: f"({t[0]} {t[1]} {t[2]})"))


@settings(**SETTINGS)
@given(_expr_strings())
def test_expr_parse_print_roundtrip(s):
    from hyperspace_amd.index.dataskipping.sketches import (
        expr_to_string, parse_expr_string)
    from hyperspace_amd.plan.expr import _expr_eq
    t1 = parse_expr_string(s)
    printed = expr_to_string(t1)
    t2 = parse_expr_string(printed)
    assert _expr_eq(t1, t2), (s, printed)


@settings(**SETTINGS)
@given(st.lists(st.integers(-10**6, 10**6), min_size=1, max_size=200),
       st.sampled_from(["a % 7", "a * 3 + 1", "(a + 5) % 11",
                        "a - 100", "a * 2 - a"]))
def test_sketch_expr_eval_matches_numpy(values, expr):
    import torch
    from hyperspace_amd.index.dataskipping.sketches import (
        eval_expr_tree, parse_expr_string)
    t = parse_expr_string(expr)
    arr = np.array(values, dtype=np.int64)
    got = eval_expr_tree(t, {"a": torch.from_numpy(arr)}).numpy()
    # Java/Spark remainder = numpy fmod (sign of the dividend), so the
    # oracle evaluates % as np.fmod
    oracle_exprs = {
        "a % 7": lambda a: np.fmod(a, 7),
        "a * 3 + 1": lambda a: a * 3 + 1,
        "(a + 5) % 11": lambda a: np.fmod(a + 5, 11),
        "a - 100": lambda a: a - 100,
        "a * 2 - a": lambda a: a * 2 - a,
    }
    assert (got == oracle_exprs[expr](arr)).all()


# ---------------------------------------------------------------------------
# murmur3 bucket stability: device-independent, null semantics
# ---------------------------------------------------------------------------

@settings(**SETTINGS)
@given(st.lists(st.integers(-2**62, 2**62), min_size=1, max_size=500),
       st.integers(min_value=1, max_value=64))
def test_bucket_ids_in_range_and_deterministic(keys, nb):
    import torch
    from hyperspace_amd.ops import cpu_ref
    t = torch.tensor(keys, dtype=torch.int64)
    b1 = cpu_ref.murmur3_bucket([t], nb)
    b2 = cpu_ref.murmur3_bucket([t], nb)
    assert (b1 == b2).all()
    assert int(b1.min()) >= 0 and int(b1.max()) < nb


# ---------------------------------------------------------------------------
# stable sort property vs numpy
# ---------------------------------------------------------------------------

@settings(**SETTINGS)
@given(st.lists(st.integers(0, 50), min_size=1, max_size=400))
def test_stable_sort_matches_numpy(keys):
    import torch
    from hyperspace_amd.ops import cpu_ref
    t = cpu_ref.normalize_key(torch.tensor(keys, dtype=torch.int64))
    payload = torch.arange(len(keys), dtype=torch.int64)
    _, perm = cpu_ref.stable_sort_u64(t, payload)
    oracle = np.argsort(np.array(keys), kind="stable")
    assert (perm.numpy() == oracle).all()


# ---------------------------------------------------------------------------
# end-to-end: random schema/config -> indexed filter == pandas oracle
# ---------------------------------------------------------------------------

@settings(max_examples=10, deadline=None)
@given(st.data())
def test_indexed_filter_matches_oracle(data):
    import tempfile
    from pathlib import Path
    rng_seed = data.draw(st.integers(0, 10**6), label="seed")
    nb = data.draw(st.sampled_from([1, 3, 8]), label="buckets")
    n = data.draw(st.sampled_from([1, 57, 2000]), label="rows")
    nullable = data.draw(st.booleans(), label="nullable")
    key_hi = data.draw(st.sampled_from([1, 10, 1000]), label="key_hi")
    op = data.draw(st.sampled_from(["=", "<", ">=", "!="]), label="op")

    rng = np.random.default_rng(rng_seed)
    tmp = Path(tempfile.mkdtemp(prefix="hs_prop"))
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp / "idx")
    d = tmp / "src"
    d.mkdir()
    key = rng.integers(0, key_hi, n)
    val = rng.random(n)
    mask = rng.random(n) > 0.2 if nullable else np.ones(n, bool)
    pq.write_table(
        pa.table({"key": pa.array(key, mask=~mask), "val": val}),
        str(d / "part-0.parquet"), compression="NONE",
        use_dictionary=False, data_page_version="1.0")
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, nb)
    h = hs.Hyperspace(session)
    df = session.read_parquet(str(d))
    h.create_index(df, hs.CoveringIndexConfig("pix", ["key"], ["val"]))
    session.enable_hyperspace()
    lit = int(rng.integers(0, max(1, key_hi)))
    out = df.filter(f"key {op} {lit}").select("key", "val").collect()

    # SQL semantics: NULL never matches any comparison (incl. !=)
    kv = key[mask]
    n_expected = int({"=": kv == lit, "<": kv < lit,
                      ">=": kv >= lit, "!=": kv != lit}[op].sum())
    assert out.num_rows == n_expected, (rng_seed, nb, n, op, lit)


# ---------------------------------------------------------------------------
# string-page decode paths vs the pyarrow oracle (dict / PLAIN /
# overflow x nullable), host reader
# ---------------------------------------------------------------------------

@settings(max_examples=15, deadline=None)
@given(
    st.lists(st.one_of(st.none(),
                       st.text(alphabet=st.characters(
                           min_codepoint=32, max_codepoint=0x2FA,
                           blacklist_characters="\x7f"),
                           max_size=12)),
             min_size=1, max_size=600),
    st.booleans(),
    st.integers(0, 2))
def test_string_pages_host_decode_matches_pyarrow(vals, use_dict, split):
    """Any mix of unicode strings and nulls, dictionary or PLAIN
    encoding, single or tiny-dict (overflow) chunks: the native host
    reader must agree with pyarrow exactly or decline (None) — never
    silently differ."""
    import tempfile
    from hyperspace_amd.execution.columnar import StringColumn
    from hyperspace_amd.sources.parquet_io import read_files_batch
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "t.parquet")
        kw = {}
        if split == 1:
            kw["dictionary_pagesize_limit"] = 512  # force dict overflow
        if split == 2:
            kw["data_page_size"] = 256  # many tiny pages
        pq.write_table(
            pa.table({"s": pa.array(vals, type=pa.string())}), p,
            compression="NONE", use_dictionary=use_dict,
            data_page_version="1.0", **kw)
        rb, counts = read_files_batch([p])
        assert counts == [len(vals)]
        col = rb.column("s")
        assert isinstance(col, StringColumn)
        got = col.to_numpy()
        m = rb.mask("s")
        for i, v in enumerate(vals):
            if v is None:
                assert m is not None and not bool(m[i])
            else:
                assert got[i] == v, (i, got[i], v)
