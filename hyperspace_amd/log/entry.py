"""Index metadata log entry model.

JSON wire format is field-for-field compatible with the reference's
``IndexLogEntry`` (index/IndexLogEntry.scala:408-590, LogEntry.scala:22-47),
including the polymorphic ``derivedDataset.type`` discriminator that Jackson
writes as the fully-qualified class name (index/Index.scala:31).  Logs written
by the reference are readable here and vice versa (for the covering kind).

Components (reference file:line):
  - FileInfo           index/IndexLogEntry.scala:40-96
  - Directory/Content  index/IndexLogEntry.scala:98-332
  - Signature/Fingerprint  index/IndexLogEntry.scala:335-343
  - Hdfs/Relation/SourcePlan/Source/Update  index/IndexLogEntry.scala:351-400
  - FileIdTracker      index/IndexLogEntry.scala:627-703
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, List, Optional, Tuple

from ..config import IndexConstants, HYPERSPACE_VERSION
from ..exceptions import HyperspaceException

LOG_VERSION = "0.1"

UNKNOWN_FILE_ID = IndexConstants.UNKNOWN_FILE_ID


# ---------------------------------------------------------------------------
# Schema (Spark StructType-compatible JSON)
# ---------------------------------------------------------------------------

_PYARROW_TO_SPARK = {
    "int64": "long", "int32": "integer", "int16": "short", "int8": "byte",
    "float": "float", "double": "double", "string": "string",
    "large_string": "string", "bool": "boolean", "date32[day]": "date",
    "timestamp[us]": "timestamp", "timestamp[ns]": "timestamp",
    "binary": "binary",
}


@dataclass
class SchemaField:
    name: str
    type: str  # spark type name: long/integer/double/string/...
    nullable: bool = True

    def to_json(self) -> Dict[str, Any]:
        return {"name": self.name, "type": self.type,
                "nullable": self.nullable, "metadata": {}}

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "SchemaField":
        return SchemaField(d["name"], d["type"], d.get("nullable", True))


@dataclass
class Schema:
    """StructType-equivalent: serialized as Spark's schema JSON."""
    fields: List[SchemaField] = field(default_factory=list)

    def to_json(self) -> Dict[str, Any]:
        return {"type": "struct", "fields": [f.to_json() for f in self.fields]}

    @staticmethod
    def from_json(d: Any) -> "Schema":
        if isinstance(d, str):
            d = json.loads(d)
        return Schema([SchemaField.from_json(f) for f in d.get("fields", [])])

    def field_names(self) -> List[str]:
        return [f.name for f in self.fields]

    def field_type(self, name: str) -> Optional[str]:
        for f in self.fields:
            if f.name.lower() == name.lower():
                return f.type
        return None

    @staticmethod
    def from_arrow(arrow_schema) -> "Schema":
        """Struct fields flatten into dotted leaf fields ("a.b.c"),
        matching the batch layer's nested flattening (the reference
        resolves nested leaves as ``__hs_nested.``-prefixed flat columns,
        util/ResolverUtils.scala)."""
        import pyarrow as pa
        fields = []

        def walk(f, prefix, nullable):
            if pa.types.is_struct(f.type):
                for child in f.type:
                    walk(child, f"{prefix}{f.name}.",
                         nullable or f.nullable)
            else:
                ts = str(f.type)
                if ts.startswith("decimal128(") or \
                        ts.startswith("decimal256("):
                    inner = ts[ts.index("(") + 1:-1].replace(" ", "")
                    t = f"decimal({inner})"
                else:
                    t = _PYARROW_TO_SPARK.get(ts, ts)
                fields.append(SchemaField(prefix + f.name, t,
                                          nullable or f.nullable))

        for f in arrow_schema:
            walk(f, "", False)
        return Schema(fields)

    def select(self, names: Iterable[str]) -> "Schema":
        lower = {f.name.lower(): f for f in self.fields}
        return Schema([lower[n.lower()] for n in names])


# ---------------------------------------------------------------------------
# Content tree
# ---------------------------------------------------------------------------

@dataclass(frozen=True)
class FileInfo:
    """(full path or basename, size, mtime-millis, stable id)."""
    name: str
    size: int
    modifiedTime: int
    id: int = UNKNOWN_FILE_ID

    def to_json(self) -> Dict[str, Any]:
        return {"name": self.name, "size": self.size,
                "modifiedTime": self.modifiedTime, "id": self.id}

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "FileInfo":
        return FileInfo(d["name"], d["size"], d["modifiedTime"],
                        d.get("id", UNKNOWN_FILE_ID))


@dataclass
class Directory:
    name: str
    files: List[FileInfo] = field(default_factory=list)
    subDirs: List["Directory"] = field(default_factory=list)

    def to_json(self) -> Dict[str, Any]:
        return {"name": self.name,
                "files": [f.to_json() for f in self.files],
                "subDirs": [d.to_json() for d in self.subDirs]}

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Directory":
        return Directory(
            d["name"],
            [FileInfo.from_json(f) for f in d.get("files", [])],
            [Directory.from_json(s) for s in d.get("subDirs", [])])

    @staticmethod
    def from_leaf_files(paths: List[Tuple[str, int, int, int]]) -> "Directory":
        """Build a Directory tree from (abspath, size, mtime, id) leaves.

        Mirrors Content.fromLeafFiles (index/IndexLogEntry.scala:289-332):
        the tree is rooted at the longest common ancestor's root.
        """
        if not paths:
            return Directory("")
        # Group by parent dir.
        by_dir: Dict[str, List[FileInfo]] = {}
        for p, size, mtime, fid in paths:
            d = os.path.dirname(p)
            by_dir.setdefault(d, []).append(
                FileInfo(os.path.basename(p), size, mtime, fid))

        # Build nested tree from root "file:/" (reference scheme-rooted
        # naming); "/" maps onto the root itself.
        root = Directory("file:/")
        nodes: Dict[str, Directory] = {"": root, "/": root}

        def get_node(dirpath: str) -> Directory:
            if dirpath in nodes:
                return nodes[dirpath]
            parent = get_node(os.path.dirname(dirpath))
            node = Directory(os.path.basename(dirpath))
            parent.subDirs.append(node)
            nodes[dirpath] = node
            return node

        for d in sorted(by_dir):
            node = get_node(d)
            node.files.extend(sorted(by_dir[d], key=lambda f: f.name))
        return root


@dataclass
class Fingerprint:
    kind: str = "NoOp"
    properties: Dict[str, Any] = field(default_factory=dict)

    def to_json(self) -> Dict[str, Any]:
        return {"kind": self.kind, "properties": self.properties}

    @staticmethod
    def from_json(d: Optional[Dict[str, Any]]) -> "Fingerprint":
        if not d:
            return Fingerprint()
        return Fingerprint(d.get("kind", "NoOp"), d.get("properties", {}))


@dataclass
class Content:
    """Directory tree + fingerprint; enumerates index data files."""
    root: Directory
    fingerprint: Fingerprint = field(default_factory=Fingerprint)

    def to_json(self) -> Dict[str, Any]:
        return {"root": self.root.to_json(),
                "fingerprint": self.fingerprint.to_json()}

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Content":
        return Content(Directory.from_json(d["root"]),
                       Fingerprint.from_json(d.get("fingerprint")))

    def files(self) -> List[str]:
        """All file paths in the tree (joined with parent names)."""
        out: List[str] = []

        def walk(node: Directory, prefix: str):
            base = _join_path(prefix, node.name)
            for f in node.files:
                out.append(_join_path(base, f.name))
            for s in node.subDirs:
                walk(s, base)

        walk(self.root, "")
        return out

    def file_infos(self) -> List[FileInfo]:
        """FileInfos with name = full path."""
        out: List[FileInfo] = []

        def walk(node: Directory, prefix: str):
            base = _join_path(prefix, node.name)
            for f in node.files:
                out.append(FileInfo(_join_path(base, f.name), f.size,
                                    f.modifiedTime, f.id))
            for s in node.subDirs:
                walk(s, base)

        walk(self.root, "")
        return out

    def os_files(self) -> List[str]:
        return [os_path(p) for p in self.files()]

    def os_file_infos(self) -> List[FileInfo]:
        return [FileInfo(os_path(f.name), f.size, f.modifiedTime, f.id)
                for f in self.file_infos()]

    @staticmethod
    def from_leaf_files(files: List[Tuple[str, int, int, int]]) -> "Content":
        return Content(Directory.from_leaf_files(files))

    @staticmethod
    def merge(a: "Content", b: "Content") -> "Content":
        """Union of two content trees (RefreshIncrementalAction merge path,
        actions/RefreshIncrementalAction.scala:115-128)."""
        leaves = {fi.name: fi for fi in a.file_infos()}
        for fi in b.file_infos():
            leaves[fi.name] = fi
        items = [(fi.name, fi.size, fi.modifiedTime, fi.id)
                 for fi in leaves.values()]
        return Content.from_leaf_files(sorted(items))


def _join_path(prefix: str, name: str) -> str:
    if not prefix:
        return name
    if prefix.endswith("/"):
        return prefix + name
    return prefix + "/" + name


def os_path(p: str) -> str:
    """Strip the 'file:' scheme from a Content path -> OS path."""
    return p[5:] if p.startswith("file:") else p


# ---------------------------------------------------------------------------
# Source description
# ---------------------------------------------------------------------------

@dataclass
class Signature:
    provider: str
    value: str

    def to_json(self):
        return {"provider": self.provider, "value": self.value}

    @staticmethod
    def from_json(d):
        return Signature(d["provider"], d["value"])


@dataclass
class LogicalPlanFingerprint:
    signatures: List[Signature] = field(default_factory=list)

    def to_json(self):
        return {"properties":
                {"signatures": [s.to_json() for s in self.signatures]},
                "kind": "LogicalPlan"}

    @staticmethod
    def from_json(d):
        sigs = d.get("properties", {}).get("signatures", [])
        return LogicalPlanFingerprint([Signature.from_json(s) for s in sigs])


@dataclass
class Update:
    """Quick-refresh delta recorded against the logged source
    (index/IndexLogEntry.scala:392-400)."""
    appendedFiles: Optional[Content] = None
    deletedFiles: Optional[Content] = None

    def to_json(self):
        return {
            "appendedFiles":
                self.appendedFiles.to_json() if self.appendedFiles else None,
            "deletedFiles":
                self.deletedFiles.to_json() if self.deletedFiles else None,
        }

    @staticmethod
    def from_json(d):
        if d is None:
            return None
        return Update(
            Content.from_json(d["appendedFiles"]) if d.get("appendedFiles")
            else None,
            Content.from_json(d["deletedFiles"]) if d.get("deletedFiles")
            else None)


@dataclass
class Hdfs:
    """Relation data: per-file content + optional update delta."""
    content: Content
    update: Optional[Update] = None

    def to_json(self):
        return {"properties": {"content": self.content.to_json(),
                               "update":
                               self.update.to_json() if self.update else None},
                "kind": "HDFS"}

    @staticmethod
    def from_json(d):
        props = d.get("properties", {})
        return Hdfs(Content.from_json(props["content"]),
                    Update.from_json(props.get("update")))


@dataclass
class Relation:
    rootPaths: List[str]
    data: Hdfs
    dataSchema: Schema
    fileFormat: str
    options: Dict[str, str] = field(default_factory=dict)

    def to_json(self):
        return {"rootPaths": self.rootPaths, "data": self.data.to_json(),
                "dataSchema": self.dataSchema.to_json(),
                "fileFormat": self.fileFormat, "options": self.options}

    @staticmethod
    def from_json(d):
        return Relation(d["rootPaths"], Hdfs.from_json(d["data"]),
                        Schema.from_json(d["dataSchema"]), d["fileFormat"],
                        d.get("options", {}))


@dataclass
class SourcePlan:
    """Mirrors the reference's SparkPlan properties wrapper
    (index/IndexLogEntry.scala:366-378); ours records the native logical
    plan's relations + fingerprint (rawPlan/sql stay null)."""
    relations: List[Relation]
    fingerprint: LogicalPlanFingerprint

    def to_json(self):
        return {"properties": {
                    "relations": [r.to_json() for r in self.relations],
                    "rawPlan": None, "sql": None,
                    "fingerprint": self.fingerprint.to_json()},
                "kind": "Spark"}

    @staticmethod
    def from_json(d):
        props = d.get("properties", {})
        return SourcePlan(
            [Relation.from_json(r) for r in props.get("relations", [])],
            LogicalPlanFingerprint.from_json(props.get("fingerprint", {})))


@dataclass
class Source:
    plan: SourcePlan

    def to_json(self):
        return {"plan": self.plan.to_json()}

    @staticmethod
    def from_json(d):
        return Source(SourcePlan.from_json(d["plan"]))


# ---------------------------------------------------------------------------
# Derived dataset (polymorphic index description)
# ---------------------------------------------------------------------------

# type-discriminator strings: we both read and write the reference's
# fully-qualified Scala class names so on-disk logs are interchangeable.
COVERING_INDEX_TYPE = "com.microsoft.hyperspace.index.covering.CoveringIndex"
ZORDER_INDEX_TYPE = (
    "com.microsoft.hyperspace.index.zordercovering.ZOrderCoveringIndex")
DATASKIPPING_INDEX_TYPE = (
    "com.microsoft.hyperspace.index.dataskipping.DataSkippingIndex")

_DERIVED_REGISTRY: Dict[str, Any] = {}


def register_derived_dataset(type_name: str, cls) -> None:
    _DERIVED_REGISTRY[type_name] = cls


def derived_dataset_from_json(d: Dict[str, Any]):
    t = d.get("type")
    cls = _DERIVED_REGISTRY.get(t)
    if cls is None:
        raise HyperspaceException(f"Unknown derivedDataset type: {t}")
    return cls.from_json(d)


# ---------------------------------------------------------------------------
# LogEntry / IndexLogEntry
# ---------------------------------------------------------------------------

@dataclass
class IndexLogEntry:
    name: str
    derivedDataset: Any  # Index instance (covering/zorder/dataskipping)
    content: Content
    source: Source
    properties: Dict[str, str] = field(default_factory=dict)
    # LogEntry base fields (index/LogEntry.scala:22-47)
    version: str = LOG_VERSION
    id: int = 0
    state: str = "UNKNOWNSTATE"
    timestamp: int = 0
    enabled: bool = True

    # -- convenience ------------------------------------------------------
    @property
    def created(self) -> bool:
        return self.state == "ACTIVE"

    @property
    def relations(self) -> List[Relation]:
        assert len(self.source.plan.relations) == 1
        return self.source.plan.relations

    def source_file_infos(self) -> List[FileInfo]:
        return self.relations[0].data.content.os_file_infos()

    def source_files_size(self) -> int:
        return sum(f.size for f in self.source_file_infos())

    def index_files_size(self) -> int:
        return sum(f.size for f in self.content.file_infos())

    @property
    def source_update(self) -> Optional[Update]:
        return self.relations[0].data.update

    def appended_files(self) -> List[FileInfo]:
        u = self.source_update
        if u and u.appendedFiles:
            return u.appendedFiles.file_infos()
        return []

    def deleted_files(self) -> List[FileInfo]:
        u = self.source_update
        if u and u.deletedFiles:
            return u.deletedFiles.file_infos()
        return []

    def has_source_update(self) -> bool:
        return bool(self.appended_files() or self.deleted_files())

    def copy_with_update(self, latest_fingerprint: LogicalPlanFingerprint,
                         appended: List[FileInfo],
                         deleted: List[FileInfo]) -> "IndexLogEntry":
        """RefreshQuickAction metadata-only delta
        (index/IndexLogEntry.scala:460-475)."""
        import copy as _copy
        e = _copy.deepcopy(self)
        rel = e.relations[0]
        rel.data.update = Update(
            appendedFiles=Content.from_leaf_files(
                [(f.name, f.size, f.modifiedTime, f.id) for f in appended])
            if appended else None,
            deletedFiles=Content.from_leaf_files(
                [(f.name, f.size, f.modifiedTime, f.id) for f in deleted])
            if deleted else None)
        e.source.plan.fingerprint = latest_fingerprint
        return e

    @property
    def signature(self) -> Optional[Signature]:
        sigs = self.source.plan.fingerprint.signatures
        return sigs[0] if sigs else None

    # -- JSON -------------------------------------------------------------
    def to_json(self) -> Dict[str, Any]:
        return {
            "name": self.name,
            "derivedDataset": self.derivedDataset.to_json(),
            "content": self.content.to_json(),
            "source": self.source.to_json(),
            "properties": self.properties,
            "version": self.version,
            "id": self.id,
            "state": self.state,
            "timestamp": self.timestamp,
            "enabled": self.enabled,
        }

    def to_json_str(self) -> str:
        # compact separators: entries with hundreds of file nodes are
        # serialized on every action write, and pretty-printing tripled
        # maintenance latency
        return json.dumps(self.to_json(), separators=(",", ":"),
                          sort_keys=False)

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "IndexLogEntry":
        if d.get("version") != LOG_VERSION:
            raise HyperspaceException(
                f"Unsupported log version {d.get('version')}")
        return IndexLogEntry(
            name=d["name"],
            derivedDataset=derived_dataset_from_json(d["derivedDataset"]),
            content=Content.from_json(d["content"]),
            source=Source.from_json(d["source"]),
            properties=d.get("properties", {}),
            version=d["version"],
            id=d.get("id", 0),
            state=d.get("state", "UNKNOWNSTATE"),
            timestamp=d.get("timestamp", 0),
            enabled=d.get("enabled", True))

    @staticmethod
    def from_json_str(s: str) -> "IndexLogEntry":
        return IndexLogEntry.from_json(json.loads(s))

    @staticmethod
    def create(name, derived, content, source, properties=None):
        props = dict(properties or {})
        props.setdefault(IndexConstants.HYPERSPACE_VERSION_PROPERTY,
                         HYPERSPACE_VERSION)
        return IndexLogEntry(name, derived, content, source, props)


# ---------------------------------------------------------------------------
# FileIdTracker
# ---------------------------------------------------------------------------

class FileIdTracker:
    """Assigns stable long ids to (path, size, mtime) triples.

    The ids are the basis of the lineage column and data-skipping per-file
    joins.  Reference: index/IndexLogEntry.scala:627-703.
    """

    def __init__(self):
        self._max_id = -1
        self._ids: Dict[Tuple[str, int, int], int] = {}

    @property
    def max_id(self) -> int:
        return self._max_id

    def add_file_info(self, files: Iterable[FileInfo]) -> None:
        for f in files:
            key = (f.name, f.size, f.modifiedTime)
            if f.id == UNKNOWN_FILE_ID:
                raise HyperspaceException(
                    f"Cannot add file info with unknown id: {f.name}")
            existing = self._ids.get(key)
            if existing is not None and existing != f.id:
                raise HyperspaceException(
                    f"Conflicting file id for {key}: {existing} vs {f.id}")
            self._ids[key] = f.id
            self._max_id = max(self._max_id, f.id)

    def add_file(self, path: str, size: int, mtime: int) -> int:
        key = (path, size, mtime)
        fid = self._ids.get(key)
        if fid is None:
            self._max_id += 1
            fid = self._max_id
            self._ids[key] = fid
        return fid

    def get_file_id(self, path: str, size: int, mtime: int) -> Optional[int]:
        return self._ids.get((path, size, mtime))

    def id_to_file_mapping(self) -> List[Tuple[int, str]]:
        return [(fid, key[0]) for key, fid in self._ids.items()]
