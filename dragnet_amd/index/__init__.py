from .sink import IndexSink  # noqa: F401
from .query import IndexQuerier  # noqa: F401
