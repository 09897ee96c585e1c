from .sketches import MinMaxSketch, BloomFilterSketch, PartitionSketch
from .index import DataSkippingIndex
from .config import DataSkippingIndexConfig
