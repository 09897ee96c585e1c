from .index import ZOrderCoveringIndex
from .config import ZOrderCoveringIndexConfig
