from .backoff import (  # noqa: F401
    PROMETHEUS_BACKOFF,
    RECONCILE_BACKOFF,
    STANDARD_BACKOFF,
    Backoff,
    retry_with_backoff,
)
from .logging import init_logger, log  # noqa: F401
