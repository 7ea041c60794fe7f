from .async_engine import AsyncEngine, worker_loop  # noqa: F401
