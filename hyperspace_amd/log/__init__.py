from .constants import States
from .entry import (
    Content, Directory, FileIdTracker, FileInfo, Fingerprint, Hdfs,
    IndexLogEntry, LogicalPlanFingerprint, Relation, Schema, SchemaField,
    Signature, Source, SourcePlan, Update, UNKNOWN_FILE_ID,
    register_derived_dataset, derived_dataset_from_json, os_path,
    COVERING_INDEX_TYPE, ZORDER_INDEX_TYPE, DATASKIPPING_INDEX_TYPE,
)
from .log_manager import IndexLogManager, IndexLogManagerFactory
from .data_manager import IndexDataManager, IndexDataManagerFactory
from .path_resolver import PathResolver
