from . import adapters, collector, constants  # noqa: F401
from .k8s import Deployment, HttpKube, InMemoryKube  # noqa: F401
from .metrics import MetricsEmitter, init_metrics  # noqa: F401
from .reconciler import Actuator, Reconciler, ReconcileResult, parse_go_duration  # noqa: F401
