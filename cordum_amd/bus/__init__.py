from .bus import Bus, LoopbackBus, RetryAfter, Subscription, compute_msg_id, subject_matches
