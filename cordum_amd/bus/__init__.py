from .bus import (
    DEFAULT_MAX_DELIVER,
    Bus,
    LoopbackBus,
    RetryAfter,
    Subscription,
    compute_msg_id,
    subject_matches,
)
