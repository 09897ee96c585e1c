"""Real Iceberg table-format conformance tests.

Pins the reader/writer to the actual HadoopTables layout the reference
consumes through the iceberg library (iceberg/IcebergRelation.scala):
vN.metadata.json + version-hint.text, manifest-list avro, manifest avro
with nested data_file records.
"""

import json
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

import hyperspace_amd as hs
from hyperspace_amd.execution.columnar import ColumnBatch
from hyperspace_amd.sources.avro_io import (read_avro_records,
                                            write_avro_records)
from hyperspace_amd.sources.iceberg_source import (IcebergTable,
                                                   IcebergTableRelation)


def _batch(k, n=400):
    return ColumnBatch({"key": torch.full((n,), k, dtype=torch.int64),
                        "val": torch.rand(n, dtype=torch.float64)})


def test_on_disk_layout(tmp_path):
    t = IcebergTable.create(str(tmp_path / "t"))
    t.append_batch(_batch(1))
    meta_dir = tmp_path / "t" / "metadata"
    names = os.listdir(meta_dir)
    assert "version-hint.text" in names
    v = int(open(meta_dir / "version-hint.text").read())
    md = json.load(open(meta_dir / f"v{v}.metadata.json"))
    assert md["format-version"] == 1
    assert "table-uuid" in md
    assert md["current-snapshot-id"] != -1
    snap = md["snapshots"][-1]
    assert snap["snapshot-id"] == md["current-snapshot-id"]
    assert snap["summary"]["operation"] == "append"
    ml = snap["manifest-list"]
    assert ml.startswith("file://") and ml.endswith(".avro")
    assert os.path.basename(ml).startswith(f"snap-{snap['snapshot-id']}-")

    # manifest list -> manifest -> data_file chain is real avro
    manifests = read_avro_records(ml[len("file://"):])
    assert manifests[0]["partition_spec_id"] == 0
    entries = read_avro_records(
        manifests[0]["manifest_path"][len("file://"):])
    assert entries[0]["status"] == 1
    df = entries[0]["data_file"]
    assert df["file_format"] == "PARQUET"
    assert df["record_count"] == 400
    assert os.path.exists(df["file_path"][len("file://"):])

    # schema carries iceberg field-ids and maps back to spark names
    sch = md["schema"]
    assert [f["name"] for f in sch["fields"]] == ["key", "val"]
    assert [f["id"] for f in sch["fields"]] == [1, 2]
    assert sch["fields"][0]["type"] == "long"
    rel = IcebergTableRelation(str(tmp_path / "t"))
    assert rel.schema.field_names() == ["key", "val"]
    assert rel.schema.field_type("val") == "double"


def test_snapshot_history_and_remove(tmp_path):
    t = IcebergTable.create(str(tmp_path / "t"))
    t.append_batch(_batch(1))
    s1 = t.snapshot_id
    t.append_batch(_batch(2))
    s2 = t.snapshot_id
    assert s1 != s2
    files_s1 = t.files_for_snapshot(s1)
    files_s2 = t.files_for_snapshot(s2)
    assert len(files_s1) == 1 and len(files_s2) == 2
    # remove one file -> manifest rewrite with EXISTING entries
    t.remove_files([files_s1[0].name])
    assert len(t.files_for_snapshot()) == 1
    # parent chain recorded
    snaps = t.snapshots()
    assert snaps[-1]["parent-snapshot-id"] == s2
    # pinned time travel unaffected by the delete
    assert len(t.files_for_snapshot(s2)) == 2


def test_reads_foreign_manifests(tmp_path):
    """Manifests written by another engine carry many extra fields
    (column bounds as maps of bytes, counts, nested unions); the generic
    avro reader must decode them and the table must plan files."""
    root = tmp_path / "ext"
    meta = root / "metadata"
    meta.mkdir(parents=True)
    datadir = root / "data"
    datadir.mkdir()
    p = datadir / "00000-0-abc.parquet"
    pq.write_table(pa.table({"key": np.arange(10)}), str(p))

    entry_schema = {
        "type": "record", "name": "manifest_entry", "fields": [
            {"name": "status", "type": "int"},
            {"name": "snapshot_id", "type": ["null", "long"],
             "default": None},
            {"name": "data_file", "type": {
                "type": "record", "name": "r2", "fields": [
                    {"name": "file_path", "type": "string"},
                    {"name": "file_format", "type": "string"},
                    {"name": "partition", "type": {
                        "type": "record", "name": "r102", "fields": []}},
                    {"name": "record_count", "type": "long"},
                    {"name": "file_size_in_bytes", "type": "long"},
                    {"name": "column_sizes", "type": ["null", {
                        "type": "map", "values": "long"}]},
                    {"name": "lower_bounds", "type": ["null", {
                        "type": "map", "values": "bytes"}]},
                    {"name": "split_offsets", "type": ["null", {
                        "type": "array", "items": "long"}]},
                ]}},
        ]}
    manifest_path = meta / "aaaa-m0.avro"
    write_avro_records(str(manifest_path), entry_schema, [{
        "status": 1, "snapshot_id": 99,
        "data_file": {
            "file_path": "file://" + str(p),
            "file_format": "PARQUET",
            "partition": {},
            "record_count": 10,
            "file_size_in_bytes": os.stat(p).st_size,
            "column_sizes": {"1": 128},
            "lower_bounds": {"1": b"\x00\x00"},
            "split_offsets": [4],
        }}])
    list_schema = {
        "type": "record", "name": "manifest_file", "fields": [
            {"name": "manifest_path", "type": "string"},
            {"name": "manifest_length", "type": "long"},
            {"name": "partition_spec_id", "type": "int"},
            {"name": "added_snapshot_id", "type": ["null", "long"]},
            {"name": "added_data_files_count", "type": ["null", "int"]},
        ]}
    ml_path = meta / "snap-99-1-bbbb.avro"
    write_avro_records(str(ml_path), list_schema, [{
        "manifest_path": "file://" + str(manifest_path),
        "manifest_length": os.stat(manifest_path).st_size,
        "partition_spec_id": 0, "added_snapshot_id": 99,
        "added_data_files_count": 1}])
    md = {
        "format-version": 2,
        "table-uuid": "cccc",
        "location": "file://" + str(root),
        "last-updated-ms": 1,
        "last-column-id": 1,
        "schemas": [{"type": "struct", "schema-id": 0, "fields": [
            {"id": 1, "name": "key", "required": False,
             "type": "long"}]}],
        "current-schema-id": 0,
        "partition-specs": [{"spec-id": 0, "fields": []}],
        "default-spec-id": 0,
        "current-snapshot-id": 99,
        "snapshots": [{"snapshot-id": 99, "timestamp-ms": 1,
                       "summary": {"operation": "append"},
                       "manifest-list": "file://" + str(ml_path),
                       "schema-id": 0}],
    }
    with open(meta / "v1.metadata.json", "w") as f:
        json.dump(md, f)
    with open(meta / "version-hint.text", "w") as f:
        f.write("1")

    t = IcebergTable(str(root))
    assert t.snapshot_id == 99
    files = t.files_for_snapshot()
    assert [os.path.basename(f.name) for f in files] == \
        ["00000-0-abc.parquet"]
    rel = IcebergTableRelation(str(root))
    assert rel.schema.field_names() == ["key"]
    assert rel.signature().startswith("99.")


def test_index_over_real_iceberg_end_to_end(tmp_path, monkeypatch):
    monkeypatch.setenv("HYPERSPACE_SYSTEM_PATH",
                       str(tmp_path / "indexes"))
    t = IcebergTable.create(str(tmp_path / "t"))
    t.append_batch(_batch(1))
    t.append_batch(_batch(2))
    session = hs.HyperspaceSession(device="cpu")
    session.conf.set(hs.IndexConstants.INDEX_NUM_BUCKETS, 4)
    h = hs.Hyperspace(session)
    df = session.read_iceberg(str(tmp_path / "t"))
    h.create_index(df, hs.CoveringIndexConfig("iix", ["key"], ["val"]))
    session.enable_hyperspace()
    assert df.filter("key = 1").collect().num_rows == 400
    snap_before = t.snapshot_id
    t.append_batch(_batch(1))
    h.refresh_index("iix", "incremental")
    df2 = session.read_iceberg(str(tmp_path / "t"))
    assert df2.filter("key = 1").collect().num_rows == 800
    # pinned snapshot query
    df_old = session.read_iceberg(str(tmp_path / "t"),
                                  snapshot_id=snap_before)
    assert df_old.filter("key = 1").collect().num_rows == 400


def test_partitioned_iceberg_identity_spec(tmp_path):
    """Identity-partitioned iceberg: partition values come from the
    manifest entries' data_file.partition record and materialize as
    constant columns; partition predicates prune before IO."""
    root = tmp_path / "pext"
    meta = root / "metadata"
    meta.mkdir(parents=True)
    rng = np.random.default_rng(3)
    rows = {}
    entries = []
    entry_schema = {
        "type": "record", "name": "manifest_entry", "fields": [
            {"name": "status", "type": "int"},
            {"name": "snapshot_id", "type": ["null", "long"]},
            {"name": "data_file", "type": {
                "type": "record", "name": "r2", "fields": [
                    {"name": "file_path", "type": "string"},
                    {"name": "file_format", "type": "string"},
                    {"name": "partition", "type": {
                        "type": "record", "name": "r102", "fields": [
                            {"name": "region",
                             "type": ["null", "string"]}]}},
                    {"name": "record_count", "type": "long"},
                    {"name": "file_size_in_bytes", "type": "long"},
                ]}},
        ]}
    for region in ("eu", "us"):
        d = root / "data" / f"region={region}"
        d.mkdir(parents=True)
        n = int(rng.integers(400, 900))
        rows[region] = n
        p = d / "00000-0.parquet"
        pq.write_table(pa.table({"key": rng.integers(0, 50, n),
                                 "v": rng.random(n)}), str(p))
        entries.append({"status": 1, "snapshot_id": 7,
                        "data_file": {
                            "file_path": "file://" + str(p),
                            "file_format": "PARQUET",
                            "partition": {"region": region},
                            "record_count": n,
                            "file_size_in_bytes": os.stat(p).st_size}})
    manifest = meta / "m0.avro"
    write_avro_records(str(manifest), entry_schema, entries)
    list_schema = {
        "type": "record", "name": "manifest_file", "fields": [
            {"name": "manifest_path", "type": "string"},
            {"name": "manifest_length", "type": "long"},
            {"name": "partition_spec_id", "type": "int"},
            {"name": "added_snapshot_id", "type": ["null", "long"]}]}
    ml = meta / "snap-7-1-x.avro"
    write_avro_records(str(ml), list_schema, [{
        "manifest_path": "file://" + str(manifest),
        "manifest_length": os.stat(manifest).st_size,
        "partition_spec_id": 0, "added_snapshot_id": 7}])
    md = {
        "format-version": 2, "table-uuid": "p1",
        "location": "file://" + str(root),
        "last-updated-ms": 1, "last-column-id": 3,
        "schemas": [{"type": "struct", "schema-id": 0, "fields": [
            {"id": 1, "name": "key", "required": False, "type": "long"},
            {"id": 2, "name": "v", "required": False, "type": "double"},
            {"id": 3, "name": "region", "required": False,
             "type": "string"}]}],
        "current-schema-id": 0,
        "partition-specs": [{"spec-id": 0, "fields": [
            {"name": "region", "transform": "identity",
             "source-id": 3, "field-id": 1000}]}],
        "default-spec-id": 0,
        "current-snapshot-id": 7,
        "snapshots": [{"snapshot-id": 7, "timestamp-ms": 1,
                       "summary": {"operation": "append"},
                       "manifest-list": "file://" + str(ml),
                       "schema-id": 0}],
    }
    with open(meta / "v1.metadata.json", "w") as f:
        json.dump(md, f)
    with open(meta / "version-hint.text", "w") as f:
        f.write("1")

    import hyperspace_amd as hs
    session = hs.HyperspaceSession(device="cpu")
    df = session.read_iceberg(str(root))
    rel = df.plan.collect_leaves()[0].relation
    assert rel.partition_schema().field_names() == ["region"]
    out = df.select("key", "region").collect()
    assert out.num_rows == sum(rows.values())
    import collections
    counts = collections.Counter(out.column("region").to_numpy().tolist())
    assert counts == rows
    from hyperspace_amd.execution.executor import Executor
    ex = Executor(session)
    got = ex.execute(df.filter("region = 'us'").select("key", "v")
                     .optimized_plan())
    assert got.num_rows == rows["us"]
    assert ex.stats.scanned_files == 1
