from .index import CoveringIndex
from .config import CoveringIndexConfig
