from .columnar import ColumnBatch, StringColumn
from .executor import Executor
