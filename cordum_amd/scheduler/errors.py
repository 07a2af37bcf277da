"""Scheduler error/reason codes (oracle: scheduler/errors.go:1-16 +
engine.go:456-472 reason-code mapping)."""
from __future__ import annotations


class SchedulingError(Exception):
    reason_code = "dispatch_failed"
    retryable = False


class NoPoolMapping(SchedulingError):
    reason_code = "no_pool_mapping"
    retryable = False


class NoWorkers(SchedulingError):
    reason_code = "no_workers"
    retryable = True


class PoolOverloaded(SchedulingError):
    reason_code = "pool_overloaded"
    retryable = True


class TenantLimit(SchedulingError):
    reason_code = "tenant_limit"
    retryable = True


REASON_MAX_RETRIES = "max_retries_exceeded"
REASON_SAFETY_DENIED = "safety_denied"
REASON_SAFETY_UNKNOWN = "safety_unknown"
REASON_DISPATCH_FAILED = "dispatch_failed"


def reason_code_for(err: Exception) -> str:
    if isinstance(err, SchedulingError):
        return err.reason_code
    return REASON_DISPATCH_FAILED


def is_retryable(err: Exception) -> bool:
    if isinstance(err, SchedulingError):
        return err.retryable
    msg = str(err).lower()
    return "no workers available" in msg or "overloaded" in msg
