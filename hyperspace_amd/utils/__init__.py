from .hashing import md5_hex
from .resolver import resolve, resolve_all
