from sentio_amd.resilience.breaker import (  # noqa: F401
    CircuitBreaker,
    CircuitOpenError,
    CircuitState,
)
from sentio_amd.resilience.retry import retry_with_backoff  # noqa: F401
from sentio_amd.resilience.fallbacks import FallbackManager, llm_fallback  # noqa: F401
from sentio_amd.resilience.health import HealthChecker, health_checker  # noqa: F401
from sentio_amd.resilience.fault_injection import (  # noqa: F401
    FaultInjector,
    InjectedFault,
)
