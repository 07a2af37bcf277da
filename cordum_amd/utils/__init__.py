from .canonical_json import canonical_json, canonical_json_hash
from .globmatch import glob_match, topic_matches
from .hashing import job_hash, sha256_hex
from .ids import new_id, new_trace_id
from .clock import Clock, SystemClock, ManualClock
