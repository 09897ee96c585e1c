from .base import Index, IndexConfigTrait, IndexerContext
