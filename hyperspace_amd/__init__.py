"""hyperspace_amd — an MI355X-native covering-index acceleration engine.

A from-scratch reimplementation of the capabilities of microsoft/hyperspace
(reference surveyed in SURVEY.md) designed MI355X-first: the index build and
query data plane runs as hand-written HIP/CDNA4 kernels over device-resident
columnar batches, scaled across GPUs with RCCL over xGMI; the metadata log,
lifecycle actions and plan-rewrite rules run on the host.

Quick start::

    import hyperspace_amd as hs
    session = hs.HyperspaceSession()
    df = session.read_parquet("/data/table")
    h = hs.Hyperspace(session)
    h.create_index(df, hs.CoveringIndexConfig("idx", ["key"], ["val"]))
    session.enable_hyperspace()
    df.filter("key = 42").collect()     # served from the index
"""

from .config import Conf, IndexConstants
from .dataframe import DataFrame
from .exceptions import HyperspaceException, KernelUnavailableError
from .hyperspace import Hyperspace
from .index.covering import CoveringIndex, CoveringIndexConfig

# backward-compat alias (reference index/package.scala:24-33:
# IndexConfig = CoveringIndexConfig)
IndexConfig = CoveringIndexConfig
from .index.dataskipping import (BloomFilterSketch, DataSkippingIndex,
                                 DataSkippingIndexConfig, MinMaxSketch,
                                 PartitionSketch)
from .index.zorder import ZOrderCoveringIndex, ZOrderCoveringIndexConfig
from .plan.expr import col, lit
from .session import HyperspaceSession, get_session, set_session

__version__ = "0.1.0"

__all__ = [
    "Conf", "IndexConstants", "DataFrame", "Hyperspace",
    "HyperspaceSession", "HyperspaceException", "KernelUnavailableError",
    "CoveringIndex", "CoveringIndexConfig", "IndexConfig", "col", "lit",
    "DataSkippingIndex", "DataSkippingIndexConfig", "MinMaxSketch",
    "BloomFilterSketch", "PartitionSketch",
    "ZOrderCoveringIndex", "ZOrderCoveringIndexConfig",
    "get_session", "set_session",
]
