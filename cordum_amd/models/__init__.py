"""Domain models — aggregated exports.

The control plane's "models" are the CAP v2 message shapes, the job state
machine, and the workflow document model. They live with their subsystems
(protocol/, workflow/, store/); this package re-exports them as the flat
model namespace mirroring the reference's model files
(core/workflow/models.go, core/protocol/pb/v1/pb.go,
core/controlplane/scheduler/types.go).
"""
from ..protocol.capv2 import (
    ActorType,
    Budget,
    BusPacket,
    ContextHints,
    DecisionType,
    Heartbeat,
    JobCancel,
    JobMetadata,
    JobPriority,
    JobProgress,
    JobRequest,
    JobResult,
    JobStatus,
    PolicyCheckRequest,
    PolicyCheckResponse,
    PolicyConstraints,
    PolicyRemediation,
    SystemAlert,
)
from ..protocol.states import JobState, TERMINAL_STATES, can_transition, is_terminal
from ..store.job_store import ApprovalRecord, SafetyDecisionRecord
from ..store.dlq_store import DLQEntry
from ..workflow.models import (
    RetryConfig,
    Step,
    StepMeta,
    StepRun,
    TimelineEvent,
    Workflow,
    WorkflowRun,
)
from ..safety.policy import (
    MCPPolicy,
    MCPRequest,
    PolicyDecision,
    PolicyInput,
    PolicyMatch,
    PolicyRule,
    SafetyPolicy,
    TenantPolicy,
)
