from .indexer import Indexer
from .common import filter_by_count

__all__ = ["Indexer", "filter_by_count"]
