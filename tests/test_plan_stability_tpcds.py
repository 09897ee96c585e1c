"""TPC-DS-shaped golden-plan suite: star schema (store_sales ⋈
date_dim ⋈ item) with covering / z-order / data-skipping / partitioned
and Delta-source variants (reference:
goldstandard/PlanStabilitySuite.scala + TPCDSBase.scala — the shipped
corpus runs TPC-DS query files over empty tables; here the rewrite
surface is exercised over small populated tables and the normalized
plans are string-compared against checked-in goldens).

Regenerate with HYPERSPACE_GENERATE_GOLDEN_FILES=1.
"""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.plan.expr import col
from test_plan_stability import GENERATE, GOLDEN_DIR, normalize  # noqa: F401
from test_plan_stability_tpch import _check


@pytest.fixture(scope="module")
def env(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("tpcds_stab")
    os.environ["HYPERSPACE_SYSTEM_PATH"] = str(tmp / "indexes")
    rng = np.random.default_rng(404)
    n = 30_000
    ss = {
        "ss_sold_date_sk": rng.integers(0, 365, n),
        "ss_item_sk": rng.integers(0, 3000, n),
        "ss_customer_sk": rng.integers(0, 5000, n),
        "ss_quantity": rng.integers(1, 100, n),
        "ss_sales_price": rng.random(n) * 200,
    }
    dd = {
        "d_date_sk": np.arange(365, dtype=np.int64),
        "d_moy": (np.arange(365) // 31 + 1).astype(np.int64),
        "d_year": np.full(365, 2003, dtype=np.int64),
    }
    it = {
        "i_item_sk": np.arange(3000, dtype=np.int64),
        "i_category": rng.integers(0, 10, 3000),
        "i_brand": rng.integers(0, 100, 3000),
    }
    # store_sales written as 10 files CLUSTERED by price/customer so
    # the data-skipping sketches can actually prune (one file per
    # price band [20i, 20i+20) and customer band [500i, 500i+500))
    d = tmp / "store_sales"
    d.mkdir()
    per = n // 10
    for i in range(10):
        sl = {k: v[i * per:(i + 1) * per].copy() for k, v in ss.items()}
        sl["ss_sales_price"] = sl["ss_sales_price"] % 20 + 20 * i
        sl["ss_customer_sk"] = sl["ss_customer_sk"] % 500 + 500 * i
        pq.write_table(pa.table(sl), str(d / f"part-{i}.parquet"))
    for name, cols in [("date_dim", dd), ("item", it)]:
        d = tmp / name
        d.mkdir()
        pq.write_table(pa.table(cols), str(d / "part-0.parquet"))
    # hive-partitioned copy of store_sales (partition col = channel)
    part_root = tmp / "store_sales_part"
    for ch in (0, 1):
        d = part_root / f"channel={ch}"
        d.mkdir(parents=True)
        sel = {k: v[ch::2] for k, v in ss.items()}
        pq.write_table(pa.table(sel), str(d / "part-0.parquet"))
    # delta-lake copy of date_dim
    from hyperspace_amd.sources.delta_source import DeltaTable
    delta_dir = tmp / "date_dim_delta"
    delta_dir.mkdir()
    pq.write_table(pa.table(dd), str(delta_dir / "part-0.parquet"))
    dt = DeltaTable.create(str(delta_dir))
    dt.append_files([str(delta_dir / "part-0.parquet")])

    # iceberg copy of item
    from hyperspace_amd.sources.iceberg_source import IcebergTable
    ice_dir = tmp / "item_ice"
    ice_dir.mkdir()
    pq.write_table(pa.table(it), str(ice_dir / "part-0.parquet"))
    ic = IcebergTable.create(str(ice_dir))
    ic.append_files([str(ice_dir / "part-0.parquet")])

    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 8)
    h = hs.Hyperspace(session)
    ssdf = session.read_parquet(str(tmp / "store_sales"))
    dddf = session.read_parquet(str(tmp / "date_dim"))
    itdf = session.read_parquet(str(tmp / "item"))
    sspart = session.read_parquet(str(part_root))
    dddelta = session.read_delta(str(delta_dir))
    itice = session.read_iceberg(str(ice_dir))
    h.create_index(ssdf, hs.CoveringIndexConfig(
        "d_ss_date", ["ss_sold_date_sk"],
        ["ss_item_sk", "ss_quantity"]))
    h.create_index(dddf, hs.CoveringIndexConfig(
        "d_dd", ["d_date_sk"], ["d_moy", "d_year"]))
    h.create_index(ssdf, hs.CoveringIndexConfig(
        "d_ss_item", ["ss_item_sk"], ["ss_quantity"]))
    h.create_index(itdf, hs.CoveringIndexConfig(
        "d_it", ["i_item_sk"], ["i_category"]))
    h.create_index(ssdf, hs.CoveringIndexConfig(
        "d_ss_qty", ["ss_quantity"], ["ss_sales_price"]))
    h.create_index(ssdf, hs.ZOrderCoveringIndexConfig(
        "d_ss_z", ["ss_item_sk", "ss_customer_sk"], ["ss_sales_price"]))
    h.create_index(ssdf, hs.DataSkippingIndexConfig(
        "d_ss_ds", hs.MinMaxSketch("ss_sales_price"),
        hs.BloomFilterSketch("ss_customer_sk", 0.01, 4096)))
    h.create_index(sspart, hs.CoveringIndexConfig(
        "d_ssp_qty", ["ss_quantity"], ["ss_sales_price"]))
    h.create_index(dddelta, hs.CoveringIndexConfig(
        "d_dd_delta", ["d_date_sk"], ["d_moy"]))
    h.create_index(itice, hs.CoveringIndexConfig(
        "d_it_ice", ["i_item_sk"], ["i_brand"]))
    session.enable_hyperspace()
    return session, ssdf, dddf, itdf, sspart, dddelta, itice


QUERIES = {
    # q3-shaped: 3-way star join — both join pairs rewritten in ONE plan
    "d01_star_join": lambda ss, dd, it, sp, dl, ii:
        ss.select("ss_sold_date_sk", "ss_item_sk", "ss_quantity")
        .join(dd.select("d_date_sk", "d_moy"),
              on=col("ss_sold_date_sk") == col("d_date_sk"))
        .join(it.select("i_item_sk", "i_category"),
              on=col("ss_item_sk") == col("i_item_sk")),
    "d02_filter_qty_eq": lambda ss, dd, it, sp, dl, ii:
        ss.filter("ss_quantity = 48")
        .select("ss_quantity", "ss_sales_price"),
    "d03_join_subset_project": lambda ss, dd, it, sp, dl, ii:
        ss.select("ss_item_sk", "ss_quantity")
        .join(it.select("i_item_sk", "i_category"),
              on=col("ss_item_sk") == col("i_item_sk")),
    # z-order: second indexed column alone still matches
    "d04_zorder_customer": lambda ss, dd, it, sp, dl, ii:
        ss.filter("ss_customer_sk = 777")
        .select("ss_customer_sk", "ss_sales_price"),
    # data-skipping: bloom eq on customer + minmax range on price
    "d05_ds_bloom_eq": lambda ss, dd, it, sp, dl, ii:
        ss.filter("ss_customer_sk = 123")
        .select("ss_sold_date_sk", "ss_sales_price"),
    "d06_ds_price_range": lambda ss, dd, it, sp, dl, ii:
        ss.filter("ss_sales_price <= 0.25")
        .select("ss_sold_date_sk", "ss_sales_price"),
    # partitioned hive source: covering rewrite over partitioned files
    "d07_partitioned_filter": lambda ss, dd, it, sp, dl, ii:
        sp.filter("ss_quantity = 11")
        .select("ss_quantity", "ss_sales_price"),
    # delta source: covering rewrite over a Delta relation
    "d08_delta_filter": lambda ss, dd, it, sp, dl, ii:
        dl.filter("d_date_sk = 100").select("d_date_sk", "d_moy"),
    # ss_customer_sk is not covered by d_ss_date -> no join rewrite
    "d09_join_not_covered": lambda ss, dd, it, sp, dl, ii:
        ss.select("ss_sold_date_sk", "ss_customer_sk")
        .join(dd.select("d_date_sk", "d_moy"),
              on=col("ss_sold_date_sk") == col("d_date_sk")),
    # IN-list + range conjunction on the filter index
    "d10_filter_in_range": lambda ss, dd, it, sp, dl, ii:
        ss.filter("ss_quantity in (5, 6, 7)")
        .filter("ss_sales_price > 100")
        .select("ss_quantity", "ss_sales_price"),
    # iceberg source: covering rewrite over an Iceberg relation
    "d11_iceberg_filter": lambda ss, dd, it, sp, dl, ii:
        ii.filter("i_item_sk = 1500").select("i_item_sk", "i_brand"),
    # fact ⋈ iceberg-dim join with both sides indexed
    "d12_join_iceberg_dim": lambda ss, dd, it, sp, dl, ii:
        ss.select("ss_item_sk", "ss_quantity")
        .join(ii.select("i_item_sk", "i_brand"),
              on=col("ss_item_sk") == col("i_item_sk")),
}


@pytest.mark.parametrize("name", sorted(QUERIES))
def test_tpcds_plan_stability(env, name):
    session, ssdf, dddf, itdf, sspart, dddelta, itice = env
    q = QUERIES[name](ssdf, dddf, itdf, sspart, dddelta, itice)
    _check(name, q.optimized_plan().pretty())
