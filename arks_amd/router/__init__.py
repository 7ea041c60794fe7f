from .app import RouterState, create_router_app  # noqa: F401
