from .models import (
    RUN_CANCELLED,
    RUN_FAILED,
    RUN_PENDING,
    RUN_RUNNING,
    RUN_SUCCEEDED,
    RUN_TERMINAL,
    RUN_TIMED_OUT,
    RUN_WAITING,
    STEP_FAILED,
    STEP_PENDING,
    STEP_RUNNING,
    STEP_SUCCEEDED,
    STEP_TERMINAL,
    STEP_WAITING,
    RetryConfig,
    Step,
    StepMeta,
    StepRun,
    TimelineEvent,
    Workflow,
    WorkflowRun,
)
from .engine import (
    Engine,
    aggregate_children,
    compute_backoff,
    deps_satisfied,
    parse_attempt,
    split_for_each_step,
    split_job_id,
    update_run_status,
)
from .eval import EvalError, eval_condition, eval_expr, eval_for_each, eval_template_string, eval_templates
from .store import RunNotFound, WorkflowNotFound, WorkflowStore
from .reconciler import RunReconciler, WorkflowService
