"""Job state machine: states, terminal set, allowed-transition table.

Semantics oracle: reference core/infra/memory/job_store.go:60-82 (terminalStates,
allowedTransitions) and core/controlplane/scheduler/types.go:61-70 (state strings).

The same table is compiled to a dense int LUT (``TRANSITION_LUT``) that the HIP
state-transition kernel (ops/hip/state_kernels.hip, K5 in SURVEY.md §2.4) loads
into constant memory: legality[from][to] as a 16x16 uint8 grid.
"""
from __future__ import annotations

from enum import IntEnum
from typing import Dict, List, Tuple


class JobState(IntEnum):
    """Dense integer encoding of job states (device representation).

    String values (wire / API representation) match the reference exactly.
    UNSPECIFIED (0) encodes the Go empty-string "no state yet" row.
    """

    UNSPECIFIED = 0
    PENDING = 1
    APPROVAL_REQUIRED = 2
    SCHEDULED = 3
    DISPATCHED = 4
    RUNNING = 5
    SUCCEEDED = 6
    FAILED = 7
    CANCELLED = 8
    TIMEOUT = 9
    DENIED = 10

    def __str__(self) -> str:  # wire form
        return "" if self is JobState.UNSPECIFIED else self.name


N_STATES = 11

_STATE_BY_NAME: Dict[str, JobState] = {s.name: s for s in JobState}
_STATE_BY_NAME[""] = JobState.UNSPECIFIED


def parse_state(s: str) -> JobState:
    try:
        return _STATE_BY_NAME[s.upper() if s else ""]
    except KeyError:
        raise ValueError(f"unknown job state {s!r}")


TERMINAL_STATES = frozenset(
    {
        JobState.SUCCEEDED,
        JobState.FAILED,
        JobState.CANCELLED,
        JobState.TIMEOUT,
        JobState.DENIED,
    }
)

# job_store.go:70-82 — first-class transition legality.
ALLOWED_TRANSITIONS: Dict[JobState, frozenset] = {
    JobState.UNSPECIFIED: frozenset(
        {
            JobState.PENDING,
            JobState.APPROVAL_REQUIRED,
            JobState.SCHEDULED,
            JobState.DISPATCHED,
            JobState.RUNNING,
            JobState.FAILED,
        }
    ),
    JobState.PENDING: frozenset(
        {
            JobState.APPROVAL_REQUIRED,
            JobState.SCHEDULED,
            JobState.DISPATCHED,
            JobState.RUNNING,
            JobState.DENIED,
            JobState.FAILED,
            JobState.TIMEOUT,
        }
    ),
    JobState.APPROVAL_REQUIRED: frozenset(
        {
            JobState.PENDING,
            JobState.SCHEDULED,
            JobState.DISPATCHED,
            JobState.RUNNING,
            JobState.DENIED,
            JobState.FAILED,
            JobState.TIMEOUT,
        }
    ),
    JobState.SCHEDULED: frozenset(
        {
            JobState.DISPATCHED,
            JobState.RUNNING,
            JobState.DENIED,
            JobState.FAILED,
            JobState.TIMEOUT,
            JobState.SUCCEEDED,
            JobState.CANCELLED,
        }
    ),
    JobState.DISPATCHED: frozenset(
        {
            JobState.RUNNING,
            JobState.SUCCEEDED,
            JobState.FAILED,
            JobState.CANCELLED,
            JobState.TIMEOUT,
        }
    ),
    JobState.RUNNING: frozenset(
        {
            JobState.SUCCEEDED,
            JobState.FAILED,
            JobState.CANCELLED,
            JobState.TIMEOUT,
        }
    ),
    JobState.SUCCEEDED: frozenset(),
    JobState.FAILED: frozenset(),
    JobState.CANCELLED: frozenset(),
    JobState.TIMEOUT: frozenset(),
    JobState.DENIED: frozenset(),
}


def is_terminal(state: JobState) -> bool:
    return state in TERMINAL_STATES


def can_transition(frm: JobState, to: JobState) -> bool:
    return to in ALLOWED_TRANSITIONS.get(frm, frozenset())


def transition_lut() -> List[List[int]]:
    """Dense legality LUT for the device kernel: lut[from][to] in {0,1}."""
    lut = [[0] * N_STATES for _ in range(N_STATES)]
    for frm, tos in ALLOWED_TRANSITIONS.items():
        for to in tos:
            lut[int(frm)][int(to)] = 1
    return lut


def transition_pairs() -> List[Tuple[int, int]]:
    return [
        (int(f), int(t)) for f, tos in ALLOWED_TRANSITIONS.items() for t in tos
    ]
