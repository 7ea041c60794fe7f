from .operator import Operator  # noqa: F401
from .store import Store  # noqa: F401
