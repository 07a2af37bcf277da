from . import capv2, subjects
from .states import (
    ALLOWED_TRANSITIONS,
    JobState,
    N_STATES,
    TERMINAL_STATES,
    can_transition,
    is_terminal,
    parse_state,
    transition_lut,
)
