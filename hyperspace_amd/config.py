"""Configuration system.

All tunables live under string keys identical to the reference's Spark conf
keys (``spark.hyperspace.*``) so that user-facing documentation carries over.
Reference: index/IndexConstants.scala:21-169 and util/HyperspaceConf.scala.

The engine-wide singleton ``conf`` can be mutated by the user; per-session
overrides are handled by ``HyperspaceContext`` holding its own ``Conf``.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional


class IndexConstants:
    # Key names (reference: index/IndexConstants.scala)
    INDEX_SYSTEM_PATH = "spark.hyperspace.system.path"
    INDEX_NUM_BUCKETS = "spark.hyperspace.index.numBuckets"
    INDEX_NUM_BUCKETS_DEFAULT = 200
    INDEX_CACHE_EXPIRY_DURATION_SECONDS = (
        "spark.hyperspace.index.cache.expiryDurationInSeconds")
    INDEX_CACHE_EXPIRY_DURATION_SECONDS_DEFAULT = 300
    INDEX_HYBRID_SCAN_ENABLED = "spark.hyperspace.index.hybridscan.enabled"
    INDEX_HYBRID_SCAN_APPENDED_RATIO_THRESHOLD = (
        "spark.hyperspace.index.hybridscan.maxAppendedRatio")
    INDEX_HYBRID_SCAN_APPENDED_RATIO_THRESHOLD_DEFAULT = 0.3
    INDEX_HYBRID_SCAN_DELETED_RATIO_THRESHOLD = (
        "spark.hyperspace.index.hybridscan.maxDeletedRatio")
    INDEX_HYBRID_SCAN_DELETED_RATIO_THRESHOLD_DEFAULT = 0.2
    INDEX_LINEAGE_ENABLED = "spark.hyperspace.index.lineage.enabled"
    INDEX_FILTER_RULE_USE_BUCKET_SPEC = (
        "spark.hyperspace.index.filterRule.useBucketSpec")
    OPTIMIZE_FILE_SIZE_THRESHOLD = (
        "spark.hyperspace.index.optimize.fileSizeThreshold")
    OPTIMIZE_FILE_SIZE_THRESHOLD_DEFAULT = 256 * 1024 * 1024
    APPLY_HYPERSPACE_ENABLED = "spark.hyperspace.apply.enabled"
    INDEX_PLAN_ANALYSIS_ENABLED = "spark.hyperspace.index.plananalysis.enabled"
    ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION = (
        "spark.hyperspace.index.zorder.targetSourceBytesPerPartition")
    ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION_DEFAULT = 1024 * 1024 * 1024
    ZORDER_QUANTILE_ENABLED = (
        "spark.hyperspace.index.zorder.quantile.enabled")
    ZORDER_QUANTILE_RELATIVE_ERROR = (
        "spark.hyperspace.index.zorder.quantile.relativeError")
    ZORDER_QUANTILE_RELATIVE_ERROR_DEFAULT = 0.01
    DATASKIPPING_TARGET_INDEX_DATA_FILE_SIZE = (
        "spark.hyperspace.index.dataskipping.targetIndexDataFileSize")
    DATASKIPPING_TARGET_INDEX_DATA_FILE_SIZE_DEFAULT = 256 * 1024 * 1024
    DATASKIPPING_MAX_INDEX_DATA_FILE_COUNT = (
        "spark.hyperspace.index.dataskipping.maxIndexDataFileCount")
    DATASKIPPING_MAX_INDEX_DATA_FILE_COUNT_DEFAULT = 10000
    DATASKIPPING_AUTO_PARTITION_SKETCH = (
        "spark.hyperspace.index.dataskipping.autoPartitionSketch")
    EVENT_LOGGER_CLASS = "spark.hyperspace.eventLoggerClass"
    DISPLAY_MODE = "spark.hyperspace.explain.displayMode"

    # Layout constants (reference: index/IndexConstants.scala:91-92)
    HYPERSPACE_LOG = "_hyperspace_log"
    INDEX_VERSION_DIRECTORY_PREFIX = "v__"

    DATA_FILE_NAME_ID_COLUMN = "_data_file_id"
    HYPERSPACE_VERSION_PROPERTY = "hyperspaceVersion"
    LINEAGE_PROPERTY = "lineage"

    UNKNOWN_FILE_ID = -1


HYPERSPACE_VERSION = "0.1.0-amd"

_DEFAULTS: Dict[str, Any] = {
    IndexConstants.INDEX_NUM_BUCKETS:
        IndexConstants.INDEX_NUM_BUCKETS_DEFAULT,
    IndexConstants.INDEX_CACHE_EXPIRY_DURATION_SECONDS:
        IndexConstants.INDEX_CACHE_EXPIRY_DURATION_SECONDS_DEFAULT,
    IndexConstants.INDEX_HYBRID_SCAN_ENABLED: False,
    IndexConstants.INDEX_HYBRID_SCAN_APPENDED_RATIO_THRESHOLD:
        IndexConstants.INDEX_HYBRID_SCAN_APPENDED_RATIO_THRESHOLD_DEFAULT,
    IndexConstants.INDEX_HYBRID_SCAN_DELETED_RATIO_THRESHOLD:
        IndexConstants.INDEX_HYBRID_SCAN_DELETED_RATIO_THRESHOLD_DEFAULT,
    IndexConstants.INDEX_LINEAGE_ENABLED: False,
    IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC: False,
    IndexConstants.OPTIMIZE_FILE_SIZE_THRESHOLD:
        IndexConstants.OPTIMIZE_FILE_SIZE_THRESHOLD_DEFAULT,
    IndexConstants.APPLY_HYPERSPACE_ENABLED: True,
    IndexConstants.INDEX_PLAN_ANALYSIS_ENABLED: False,
    IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION:
        IndexConstants.ZORDER_TARGET_SOURCE_BYTES_PER_PARTITION_DEFAULT,
    IndexConstants.ZORDER_QUANTILE_ENABLED: False,
    IndexConstants.ZORDER_QUANTILE_RELATIVE_ERROR:
        IndexConstants.ZORDER_QUANTILE_RELATIVE_ERROR_DEFAULT,
    IndexConstants.DATASKIPPING_TARGET_INDEX_DATA_FILE_SIZE:
        IndexConstants.DATASKIPPING_TARGET_INDEX_DATA_FILE_SIZE_DEFAULT,
    IndexConstants.DATASKIPPING_MAX_INDEX_DATA_FILE_COUNT:
        IndexConstants.DATASKIPPING_MAX_INDEX_DATA_FILE_COUNT_DEFAULT,
    IndexConstants.DATASKIPPING_AUTO_PARTITION_SKETCH: True,
    IndexConstants.EVENT_LOGGER_CLASS: None,
    IndexConstants.DISPLAY_MODE: "plaintext",
    IndexConstants.INDEX_SYSTEM_PATH: None,
}


class Conf:
    """Typed accessor over a string-keyed config dict.

    Reference: util/HyperspaceConf.scala:27-238.
    """

    def __init__(self, overrides: Optional[Dict[str, Any]] = None):
        self._values: Dict[str, Any] = {}
        if overrides:
            self._values.update(overrides)

    def get(self, key: str, default: Any = None) -> Any:
        if key in self._values:
            return self._values[key]
        if key in _DEFAULTS:
            return _DEFAULTS[key]
        return default

    def set(self, key: str, value: Any) -> None:
        self._values[key] = value

    def unset(self, key: str) -> None:
        self._values.pop(key, None)

    def copy(self) -> "Conf":
        return Conf(copy.deepcopy(self._values))

    # --- typed accessors -------------------------------------------------
    @property
    def num_buckets(self) -> int:
        return int(self.get(IndexConstants.INDEX_NUM_BUCKETS))

    @property
    def system_path(self) -> Optional[str]:
        return self.get(IndexConstants.INDEX_SYSTEM_PATH)

    @property
    def hybrid_scan_enabled(self) -> bool:
        return bool(self.get(IndexConstants.INDEX_HYBRID_SCAN_ENABLED))

    @property
    def hybrid_scan_appended_ratio_threshold(self) -> float:
        return float(
            self.get(IndexConstants.INDEX_HYBRID_SCAN_APPENDED_RATIO_THRESHOLD))

    @property
    def hybrid_scan_deleted_ratio_threshold(self) -> float:
        return float(
            self.get(IndexConstants.INDEX_HYBRID_SCAN_DELETED_RATIO_THRESHOLD))

    @property
    def lineage_enabled(self) -> bool:
        return bool(self.get(IndexConstants.INDEX_LINEAGE_ENABLED))

    @property
    def filter_rule_use_bucket_spec(self) -> bool:
        return bool(self.get(IndexConstants.INDEX_FILTER_RULE_USE_BUCKET_SPEC))

    @property
    def optimize_file_size_threshold(self) -> int:
        return int(self.get(IndexConstants.OPTIMIZE_FILE_SIZE_THRESHOLD))

    @property
    def apply_enabled(self) -> bool:
        return bool(self.get(IndexConstants.APPLY_HYPERSPACE_ENABLED))

    @property
    def plan_analysis_enabled(self) -> bool:
        return bool(self.get(IndexConstants.INDEX_PLAN_ANALYSIS_ENABLED))

    @property
    def cache_expiry_seconds(self) -> int:
        return int(
            self.get(IndexConstants.INDEX_CACHE_EXPIRY_DURATION_SECONDS))
