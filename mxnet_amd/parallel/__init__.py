from . import kvstore  # noqa: F401
from .kvstore import create, KVStore, KVStoreBase, DistKVStore  # noqa: F401
