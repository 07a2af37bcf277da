from .job_store import (
    ApprovalRecord,
    InvalidTransition,
    JobStore,
    SafetyDecisionRecord,
    ACTIVE_STATES,
)
from .memory_store import MemoryStore, key_from_pointer, pointer_for_key
from .dlq_store import DLQEntry, DLQStore
from .locks import LockService, MODE_EXCLUSIVE, MODE_SHARED
from .artifacts import ArtifactStore
from .schema_registry import SchemaRegistry, SchemaValidationError, validate_value
from .configsvc import ConfigService, EffectiveSnapshot, json_merge_patch
