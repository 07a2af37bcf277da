from .engine import Engine, Metrics, apply_constraints, QUEUE_GROUP
from .errors import (
    NoPoolMapping,
    NoWorkers,
    PoolOverloaded,
    SchedulingError,
    TenantLimit,
    is_retryable,
    reason_code_for,
)
from .registry import WorkerRegistry
from .reconciler import PendingReplayer, Reconciler
from .safety_client import SafetyChecker, build_check_request, extract_tenant
from .strategy import (
    LeastLoadedStrategy,
    NaiveStrategy,
    PoolProfile,
    PoolRouting,
    Strategy,
    load_score,
    is_overloaded,
    routing_from_pools_yaml,
)
