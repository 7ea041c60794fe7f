from .app import BackendResolver, create_gateway_app  # noqa: F401
from .limiter import InMemoryCounterStore, LimitDescriptor, RateLimiter  # noqa: F401
from .provider import ConfigProvider, UserQos  # noqa: F401
from .quota import QuotaDescriptor, QuotaService  # noqa: F401
