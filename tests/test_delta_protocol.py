"""Real Delta Lake protocol conformance tests.

The reference consumes actual Delta transaction logs via
TahoeLogFileIndex (delta/DeltaLakeRelation.scala:33-44); these tests
pin our reader/writer to the open Delta protocol: 20-digit NDJSON
commits, single-key actions, url-encoded relative paths, metaData
schemaString, parquet checkpoints + _last_checkpoint.
"""

import json
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest

import hyperspace_amd as hs
from hyperspace_amd.sources.delta_source import (DeltaTable,
                                                 DeltaTableRelation)


def _write_part(d, rng, n=1000, name="f0.parquet"):
    p = os.path.join(d, name)
    pq.write_table(pa.table({"key": rng.integers(0, 100, n),
                             "val": rng.random(n)}), p)
    return p


def test_on_disk_commit_format(tmp_path):
    """v0 must carry protocol + metaData; adds are NDJSON single-key
    actions with url-encoded relative paths."""
    rng = np.random.default_rng(1)
    t = DeltaTable.create(str(tmp_path / "t"))
    sub = tmp_path / "t" / "sub dir"
    sub.mkdir()
    p = _write_part(str(sub), rng, name="a b.parquet")
    t.append_files([p])

    log = tmp_path / "t" / "_delta_log"
    names = sorted(os.listdir(log))
    assert f"{0:020d}.json" in names and f"{1:020d}.json" in names

    v0 = [json.loads(l) for l in open(log / f"{0:020d}.json")]
    assert all(len(a) == 1 for a in v0)
    keys = {next(iter(a)) for a in v0}
    assert {"protocol", "metaData"} <= keys
    proto = next(a["protocol"] for a in v0 if "protocol" in a)
    assert proto["minReaderVersion"] == 1

    v1 = [json.loads(l) for l in open(log / f"{1:020d}.json")]
    adds = [a["add"] for a in v1 if "add" in a]
    assert len(adds) == 1
    assert adds[0]["path"] == "sub%20dir/a%20b.parquet"  # encoded, rel
    assert adds[0]["dataChange"] is True
    assert adds[0]["size"] == os.stat(p).st_size
    # metaData schemaString recorded on first append (Spark struct JSON)
    md = [a["metaData"] for a in v1 if "metaData" in a]
    ss = json.loads(md[0]["schemaString"])
    assert ss["type"] == "struct"
    assert [f["name"] for f in ss["fields"]] == ["key", "val"]

    files = t.files_at()
    assert [os.path.basename(f.name) for f in files] == ["a b.parquet"]


def test_reads_foreign_log(tmp_path):
    """A log written by another engine (delta-rs/Spark style: commitInfo
    variants, stats/tags fields, unknown action keys) must replay."""
    root = tmp_path / "ext"
    log = root / "_delta_log"
    log.mkdir(parents=True)
    rng = np.random.default_rng(5)
    _write_part(str(root), rng, name="part-00000.parquet")
    _write_part(str(root), rng, name="part-00001.parquet")
    schema_string = json.dumps(
        {"type": "struct", "fields": [
            {"name": "key", "type": "long", "nullable": True,
             "metadata": {}},
            {"name": "val", "type": "double", "nullable": True,
             "metadata": {}}]})
    with open(log / f"{0:020d}.json", "w") as f:
        f.write(json.dumps({"protocol": {"minReaderVersion": 1,
                                         "minWriterVersion": 2}}) + "\n")
        f.write(json.dumps({"metaData": {
            "id": "11111111-2222-3333-4444-555555555555",
            "format": {"provider": "parquet", "options": {}},
            "schemaString": schema_string,
            "partitionColumns": [], "configuration": {},
            "createdTime": 170000}}) + "\n")
        f.write(json.dumps({"add": {
            "path": "part-00000.parquet", "partitionValues": {},
            "size": os.stat(root / "part-00000.parquet").st_size,
            "modificationTime": 170001, "dataChange": True,
            "stats": "{\"numRecords\":1000}",
            "tags": {"INSERTION_TIME": "170001"}}}) + "\n")
        f.write(json.dumps({"commitInfo": {"operation": "WRITE"}}) + "\n")
    with open(log / f"{1:020d}.json", "w") as f:
        f.write(json.dumps({"add": {
            "path": "part-00001.parquet", "partitionValues": {},
            "size": os.stat(root / "part-00001.parquet").st_size,
            "modificationTime": 170002, "dataChange": True}}) + "\n")
        f.write(json.dumps({"remove": {
            "path": "part-00000.parquet", "deletionTimestamp": 170003,
            "dataChange": True}}) + "\n")
        # unknown action types must be ignored, not crash
        f.write(json.dumps({"txn": {"appId": "x", "version": 3}}) + "\n")

    t = DeltaTable(str(root))
    assert t.version == 1
    v0_files = [os.path.basename(f.name) for f in t.files_at(0)]
    assert v0_files == ["part-00000.parquet"]
    v1_files = [os.path.basename(f.name) for f in t.files_at(1)]
    assert v1_files == ["part-00001.parquet"]
    # schema comes from metaData.schemaString, not a parquet footer
    rel = DeltaTableRelation(str(root))
    assert rel.schema.field_names() == ["key", "val"]
    assert rel.schema.field_type("key") == "long"


def test_checkpoint_roundtrip_and_log_cleanup(tmp_path):
    """After CHECKPOINT_INTERVAL commits a checkpoint parquet is written;
    snapshots must reconstruct from checkpoint + later commits even when
    earlier JSON commits are deleted (log retention)."""
    rng = np.random.default_rng(7)
    t = DeltaTable.create(str(tmp_path / "t"))
    paths = []
    for i in range(12):
        p = _write_part(str(tmp_path / "t"), rng, n=50,
                        name=f"p{i:02d}.parquet")
        paths.append(p)
        t.append_files([p])
    t.remove_files([paths[0]])  # v13
    log = tmp_path / "t" / "_delta_log"
    assert (log / f"{10:020d}.checkpoint.parquet").exists()
    lc = json.load(open(log / "_last_checkpoint"))
    assert lc["version"] == 10

    # drop pre-checkpoint JSON commits: replay must use the checkpoint
    t.clean_commits_before(10)
    assert not (log / f"{0:020d}.json").exists()
    t2 = DeltaTable(str(tmp_path / "t"))
    names = [os.path.basename(f.name) for f in t2.files_at()]
    assert names == [f"p{i:02d}.parquet" for i in range(1, 12)]
    # time travel to a post-checkpoint version still works
    names11 = [os.path.basename(f.name) for f in t2.files_at(11)]
    assert names11 == [f"p{i:02d}.parquet" for i in range(11)]
    # checkpoint-version snapshot itself
    names10 = [os.path.basename(f.name) for f in t2.files_at(10)]
    assert names10 == [f"p{i:02d}.parquet" for i in range(10)]


def test_checkpoint_readable_by_pyarrow_with_map_partition_values(
        tmp_path):
    """The checkpoint we write uses the spec's map<string,string>
    partitionValues; a foreign checkpoint with values must replay."""
    rng = np.random.default_rng(9)
    t = DeltaTable.create(str(tmp_path / "t"))
    p = _write_part(str(tmp_path / "t"), rng)
    t.append_files([p])
    out = t.checkpoint()
    table = pq.read_table(out)
    assert {"protocol", "metaData", "add", "remove"} <= set(
        table.column_names)
    adds = [r["add"] for r in table.to_pylist() if r["add"] is not None]
    assert len(adds) == 1 and adds[0]["path"] == "f0.parquet"


def test_index_over_real_delta_end_to_end(tmp_path, monkeypatch):
    """createIndex + commit + incremental refresh + time travel over the
    real-protocol table (reference DeltaLakeIntegrationTest behavior)."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH",
                       str(tmp_path / "indexes"))
    rng = np.random.default_rng(21)
    t = DeltaTable.create(str(tmp_path / "t"))
    from hyperspace_amd.execution.columnar import ColumnBatch
    import torch

    def batch(k):
        return ColumnBatch({
            "key": torch.full((500,), k, dtype=torch.int64),
            "val": torch.rand(500, dtype=torch.float64)})

    t.append_batch(batch(1))  # v1
    t.append_batch(batch(2))  # v2
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_delta(str(tmp_path / "t"))
    h.create_index(df, hs.CoveringIndexConfig("dix", ["key"], ["val"]))
    session.enable_hyperspace()
    assert df.filter("key = 1").collect().num_rows == 500

    t.append_batch(batch(1))  # v3: +500 rows of key=1
    h.refresh_index("dix", "incremental")
    df2 = session.read_delta(str(tmp_path / "t"))
    assert df2.filter("key = 1").collect().num_rows == 1000
    # pinned time travel still sees the old snapshot
    df_v2 = session.read_delta(str(tmp_path / "t"), version_as_of=2)
    assert df_v2.filter("key = 1").collect().num_rows == 500


def test_partitioned_delta_table(tmp_path, monkeypatch):
    """Partitioned Delta: partition columns come from
    metaData.partitionColumns + each add's partitionValues (NOT from the
    data files); queries project and prune on them
    (reference DeltaLakeRelation partition handling)."""
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(31)
    from hyperspace_amd.log.entry import Schema, SchemaField
    schema = Schema([SchemaField("key", "long", True),
                     SchemaField("val", "double", True),
                     SchemaField("region", "string", False)])
    t = DeltaTable.create(str(tmp_path / "t"), schema=schema,
                          partition_columns=["region"])
    pv_map = {}
    per_region_rows = {}
    for region in ("eu", "us", "ap"):
        d = tmp_path / "t" / f"region={region}"
        d.mkdir()
        n = int(rng.integers(500, 1500))
        per_region_rows[region] = n
        p = d / "part-0.parquet"
        pq.write_table(pa.table({"key": rng.integers(0, 50, n),
                                 "val": rng.random(n)}), str(p))
        pv_map[str(p)] = {"region": region}
    t.append_files(list(pv_map), partition_values=pv_map)

    import hyperspace_amd as hs
    session = hs.HyperspaceSession(device="cpu")
    h = hs.Hyperspace(session)
    df = session.read_delta(str(tmp_path / "t"))
    # schema includes the partition column with its declared type
    scan = df.plan.collect_leaves()[0]
    assert scan.relation.schema.field_type("region") == "string"
    assert scan.relation.partition_schema().field_names() == ["region"]

    # projection materializes the partition constants
    out = df.select("key", "region").collect()
    assert out.num_rows == sum(per_region_rows.values())
    vals = out.column("region").to_numpy()
    import collections
    counts = collections.Counter(vals.tolist())
    assert counts == per_region_rows

    # partition-only predicate prunes to the one file before IO
    from hyperspace_amd.execution.executor import Executor
    ex = Executor(session)
    q = df.filter("region = 'us'").select("key", "val", "region")
    got = ex.execute(q.optimized_plan())
    assert got.num_rows == per_region_rows["us"]
    assert ex.stats.scanned_files == 1

    # covering index over a partitioned delta table still answers
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h.create_index(df, hs.CoveringIndexConfig("pdx", ["key"], ["val"]))
    session.enable_hyperspace()
    n7 = df.filter("key = 7").select("key", "val").collect().num_rows
    session.disable_hyperspace()
    assert n7 == df.filter("key = 7").collect().num_rows


def test_timestamp_as_of_time_travel(tmp_path, monkeypatch):
    """timestampAsOf resolves the newest version committed at or before
    the given time (Delta semantics; commitInfo timestamps)."""
    import json as _json
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH", str(tmp_path / "idx"))
    rng = np.random.default_rng(3)
    t = DeltaTable.create(str(tmp_path / "t"))
    p1 = _write_part(str(tmp_path / "t"), rng, name="a.parquet")
    t.append_files([p1])  # v1
    p2 = _write_part(str(tmp_path / "t"), rng, name="b.parquet")
    t.append_files([p2])  # v2
    # rewrite commit timestamps to known values
    log = tmp_path / "t" / "_delta_log"
    for v, ts in ((0, 1_000_000), (1, 2_000_000), (2, 3_000_000)):
        pth = log / f"{v:020d}.json"
        lines = [ _json.loads(l) for l in open(pth) ]
        for a in lines:
            if "commitInfo" in a:
                a["commitInfo"]["timestamp"] = ts
        with open(pth, "w") as f:
            for a in lines:
                f.write(_json.dumps(a) + "\n")

    assert t.version_at_timestamp(2_500_000) == 1
    assert t.version_at_timestamp(3_000_000) == 2
    with pytest.raises(Exception, match="No delta version"):
        t.version_at_timestamp(500_000)

    import hyperspace_amd as hs
    session = hs.HyperspaceSession(device="cpu")
    df = session.read_delta(str(tmp_path / "t"),
                            timestamp_as_of=2_500_000)
    assert df.collect().num_rows == 1000  # only a.parquet at v1
    df2 = session.read_delta(str(tmp_path / "t"),
                             timestamp_as_of=3_500_000)
    assert df2.collect().num_rows == 2000
    with pytest.raises(Exception, match="only one of"):
        session.read_delta(str(tmp_path / "t"), version_as_of=1,
                           timestamp_as_of=2_500_000)
