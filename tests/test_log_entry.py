"""IndexLogEntry JSON round-trip tests.

Mirrors the reference's IndexLogEntryTest.scala (JSON wire-format
stability) — field names and nesting must match the reference schema.
"""

import json

from hyperspace_amd.log import (
    Content, Directory, FileIdTracker, FileInfo, Hdfs, IndexLogEntry,
    LogicalPlanFingerprint, Relation, Schema, SchemaField, Signature,
    Source, SourcePlan, Update, COVERING_INDEX_TYPE)
from hyperspace_amd.index.covering import CoveringIndex


def make_entry():
    schema = Schema([SchemaField("RGUID", "string"),
                     SchemaField("Date", "string")])
    source_content = Content(Directory(
        "test", [FileInfo("f1", 100, 100, 0), FileInfo("f2", 100, 200, 1)]))
    relation = Relation(
        rootPaths=["rootpath"],
        data=Hdfs(source_content,
                  Update(None, Content(Directory(
                      "", [FileInfo("f1", 10, 10, 2)])))),
        dataSchema=schema,
        fileFormat="parquet",
        options={})
    plan = SourcePlan([relation],
                      LogicalPlanFingerprint(
                          [Signature("provider", "signatureValue")]))
    entry = IndexLogEntry.create(
        "indexName",
        CoveringIndex(["col1"], ["col2", "col3"], schema, 200, {}),
        Content(Directory("rootContentPath")),
        Source(plan),
        {})
    entry.state = "ACTIVE"
    entry.timestamp = 1578818514080
    return entry


def test_json_roundtrip():
    entry = make_entry()
    s = entry.to_json_str()
    back = IndexLogEntry.from_json_str(s)
    assert back.name == "indexName"
    assert back.state == "ACTIVE"
    assert back.derivedDataset.indexed_columns == ["col1"]
    assert back.derivedDataset.included_columns == ["col2", "col3"]
    assert back.derivedDataset.num_buckets == 200
    assert back.to_json() == entry.to_json()


def test_wire_format_fields():
    d = make_entry().to_json()
    # top-level envelope (reference: LogEntry.scala)
    for key in ("name", "derivedDataset", "content", "source", "properties",
                "version", "id", "state", "timestamp", "enabled"):
        assert key in d
    # polymorphic discriminator uses the reference's class name
    assert d["derivedDataset"]["type"] == COVERING_INDEX_TYPE
    assert d["derivedDataset"]["indexedColumns"] == ["col1"]
    assert d["derivedDataset"]["includedColumns"] == ["col2", "col3"]
    assert d["derivedDataset"]["numBuckets"] == 200
    # schema is Spark StructType JSON
    assert d["derivedDataset"]["schema"]["type"] == "struct"
    # source nesting
    rel = d["source"]["plan"]["properties"]["relations"][0]
    assert rel["rootPaths"] == ["rootpath"]
    assert rel["data"]["kind"] == "HDFS"
    files = rel["data"]["properties"]["content"]["root"]["files"]
    assert files[0] == {"name": "f1", "size": 100, "modifiedTime": 100,
                        "id": 0}
    assert d["source"]["plan"]["kind"] == "Spark"
    fp = d["source"]["plan"]["properties"]["fingerprint"]
    assert fp["kind"] == "LogicalPlan"
    assert fp["properties"]["signatures"][0]["provider"] == "provider"


def test_content_files():
    c = Content(Directory("file:/", [], [
        Directory("a", [FileInfo("f1", 0, 0, -1), FileInfo("f2", 0, 0, -1)],
                  [Directory("b", [FileInfo("f3", 0, 0, -1)])])]))
    assert set(c.files()) == {"file:/a/f1", "file:/a/f2", "file:/a/b/f3"}


def test_content_from_leaf_files_roundtrip():
    files = [("/data/x/part-0.parquet", 10, 1, 0),
             ("/data/x/part-1.parquet", 20, 2, 1),
             ("/data/y/part-2.parquet", 30, 3, 2)]
    c = Content.from_leaf_files(files)
    infos = {f.name: (f.size, f.modifiedTime, f.id) for f in c.file_infos()}
    assert infos == {
        "file:/data/x/part-0.parquet": (10, 1, 0),
        "file:/data/x/part-1.parquet": (20, 2, 1),
        "file:/data/y/part-2.parquet": (30, 3, 2)}


def test_content_merge():
    a = Content.from_leaf_files([("/d/f1", 1, 1, 0)])
    b = Content.from_leaf_files([("/d/f2", 2, 2, 1)])
    merged = Content.merge(a, b)
    assert len(merged.file_infos()) == 2


def test_file_id_tracker():
    t = FileIdTracker()
    id1 = t.add_file("/a/f1", 10, 100)
    id2 = t.add_file("/a/f2", 10, 100)
    assert (id1, id2) == (0, 1)
    # same triple -> same id
    assert t.add_file("/a/f1", 10, 100) == 0
    # changed size -> new id
    assert t.add_file("/a/f1", 11, 100) == 2
    assert t.max_id == 2


def test_copy_with_update():
    entry = make_entry()
    updated = entry.copy_with_update(
        LogicalPlanFingerprint([Signature("p", "v2")]),
        appended=[FileInfo("/d/f9", 5, 5, 9)],
        deleted=[])
    assert [f.id for f in updated.appended_files()] == [9]
    assert entry.appended_files() == []  # original untouched
    assert updated.signature.value == "v2"


def test_cache_with_transform():
    from hyperspace_amd.utils.cache import CacheWithTransform
    calls = []
    state = {"k": 1}
    c = CacheWithTransform(lambda: state["k"], lambda k: calls.append(k)
                           or k * 10)
    assert c.load() == 10
    assert c.load() == 10
    assert calls == [1]       # memoized
    state["k"] = 2
    assert c.load() == 20     # conf change invalidates
    assert calls == [1, 2]


def test_covering_config_builder():
    """Builder pattern parity (reference CoveringIndexConfig.Builder)."""
    import pytest as _pytest
    import hyperspace_amd as hs
    from hyperspace_amd.exceptions import HyperspaceException
    cfg = (hs.CoveringIndexConfig.builder()
           .index_name("bix").index_by("k1", "k2").include("v").create())
    assert cfg.index_name == "bix"
    assert cfg.indexed_columns == ["k1", "k2"]
    assert cfg.included_columns == ["v"]
    with _pytest.raises(HyperspaceException):
        hs.CoveringIndexConfig.builder().index_name("a").index_name("b")
    with _pytest.raises(HyperspaceException):
        hs.CoveringIndexConfig.builder().index_by("x").create()  # no name
    # camelCase aliases
    cfg2 = (hs.CoveringIndexConfig.builder()
            .indexName("c").indexBy("k").create())
    assert cfg2.indexed_columns == ["k"]
