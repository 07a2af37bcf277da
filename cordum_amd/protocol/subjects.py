"""Bus subject constants (wire-compatible with the reference's CAP SDK).

Reference: core/protocol/capsdk/constants.go:1-14. Subject → native-transport
mapping (RCCL/xGMI) is documented in SURVEY.md §2.5.
"""

SUBJECT_SUBMIT = "sys.job.submit"
SUBJECT_RESULT = "sys.job.result"
SUBJECT_HEARTBEAT = "sys.heartbeat"
SUBJECT_PROGRESS = "sys.job.progress"
SUBJECT_CANCEL = "sys.job.cancel"
SUBJECT_DLQ = "sys.job.dlq"
SUBJECT_WORKFLOW_EVENT = "sys.workflow.event"

DEFAULT_PROTOCOL_VERSION = 1


def worker_subject(worker_id: str) -> str:
    """Direct per-worker subject (bus/nats.go:94-99)."""
    return f"worker.{worker_id}.jobs"


def pool_subject(topic: str) -> str:
    """Pool subjects are the topic itself (job.*)."""
    return topic


# Durable subjects get at-least-once + msg-id dedup semantics on the native
# bus (reference: infra/bus/nats.go:369-403 isDurableSubject).
def is_durable_subject(subject: str) -> bool:
    if subject in (SUBJECT_SUBMIT, SUBJECT_RESULT, SUBJECT_DLQ):
        return True
    if subject.startswith("job."):
        return True
    if subject.startswith("worker.") and subject.endswith(".jobs"):
        return True
    return False
