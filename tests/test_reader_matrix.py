"""Native-reader conformance matrix: codec x encoding x page version
x nullability x column type against the pyarrow oracle.

CPU half: uncompressed combos decode through read_files_batch (host
native path).  GPU half: every combo (incl. codecs) through
read_files_batch_device.  The native layout must either decode to
EXACTLY the pyarrow content or decline (fall back) — never silently
differ.  (Reference reads all shapes through parquet-mr; SURVEY §2.6
K1.)
"""

import itertools
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from hyperspace_amd.execution.columnar import StringColumn
from hyperspace_amd.sources.parquet_io import (read_files_batch,
                                               read_files_batch_device)

N = 60_000


def _table(with_nulls):
    rng = np.random.default_rng(9)
    cols = {
        "i64": pa.array(rng.integers(-10**12, 10**12, N)),
        "i32": pa.array(rng.integers(-10**6, 10**6, N).astype(np.int32)),
        "f64": pa.array(rng.random(N)),
        "s": pa.array([f"v{i % 499:03d}" for i in range(N)],
                      type=pa.string()),
    }
    if with_nulls:
        cols["ni"] = pa.array([None if i % 7 == 0 else int(i)
                               for i in range(N)], type=pa.int64())
        cols["ns"] = pa.array([None if i % 5 == 0 else f"x{i % 31}"
                               for i in range(N)], type=pa.string())
    return pa.table(cols)


def _write(tmp_path, codec, use_dict, dpv, with_nulls):
    p = str(tmp_path / f"m_{codec}_{use_dict}_{dpv}_{with_nulls}.parquet")
    pq.write_table(_table(with_nulls), p, compression=codec,
                   use_dictionary=use_dict, data_page_version=dpv,
                   row_group_size=25_000)
    return p


def _assert_matches(batch, table, rc):
    assert rc == [N]
    for name in table.column_names:
        col = table.column(name)
        if pa.types.is_string(col.type):
            s = batch.column(name)
            assert isinstance(s, StringColumn)
            codes = s.codes.cpu().numpy()
            m = batch.mask(name)
            mv = (m.cpu().numpy() if m is not None
                  else np.ones(N, dtype=bool))
            vals = col.to_pylist()
            for i in range(0, N, 313):
                if vals[i] is None:
                    assert not mv[i], (name, i)
                else:
                    assert s.values[codes[i]] == vals[i], (name, i)
        else:
            got = batch.tensor(name).cpu().numpy()
            m = batch.mask(name)
            ref = col.to_numpy(zero_copy_only=False)
            if m is None:
                assert np.allclose(got, ref), name
            else:
                mv = m.cpu().numpy()
                ok = ~np.isnan(ref.astype(np.float64)) \
                    if ref.dtype.kind == "f" else ~pa.compute.is_null(
                        col).to_numpy(zero_copy_only=False)
                assert (mv == ok).all(), name
                assert np.allclose(got[mv], ref[mv]), name


CPU_COMBOS = list(itertools.product(["NONE"], [True, False],
                                    ["1.0", "2.0"], [True, False]))


@pytest.mark.parametrize("codec,use_dict,dpv,with_nulls", CPU_COMBOS)
def test_reader_matrix_cpu(tmp_path, codec, use_dict, dpv, with_nulls):
    p = _write(tmp_path, codec, use_dict, dpv, with_nulls)
    t = _table(with_nulls)
    batch, rc = read_files_batch([p])
    _assert_matches(batch, t, rc)


GPU_COMBOS = list(itertools.product(
    ["NONE", "SNAPPY", "ZSTD", "GZIP", "LZ4"], [True, False],
    ["1.0", "2.0"], [True, False]))


@pytest.mark.gpu
@pytest.mark.parametrize("codec,use_dict,dpv,with_nulls", GPU_COMBOS)
def test_reader_matrix_gpu(tmp_path, codec, use_dict, dpv, with_nulls):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    p = _write(tmp_path, codec, use_dict, dpv, with_nulls)
    t = _table(with_nulls)
    batch, rc = read_files_batch_device([p], torch.device("cuda:0"))
    _assert_matches(batch, t, rc)
