from .aggregator import Aggregator  # noqa: F401
from .flat import FlatParams  # noqa: F401
