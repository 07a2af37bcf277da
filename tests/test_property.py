"""Property-based tests (hypothesis): codec round-trips and compiled-policy
agreement on adversarial inputs."""
from hypothesis import given, settings, strategies as st

from cordum_amd.protocol.capv2 import Budget, JobMetadata, JobRequest, JobPriority
from cordum_amd.utils.globmatch import glob_match
from cordum_amd.utils.hashing import job_hash

label_keys = st.text(alphabet=st.characters(blacklist_categories=("Cs",), max_codepoint=0x2FF),
                     min_size=1, max_size=12)
texts = st.text(alphabet=st.characters(blacklist_categories=("Cs",), max_codepoint=0x2FF), max_size=24)


@settings(max_examples=200, deadline=None)
@given(
    job_id=texts, topic=texts, tenant=texts,
    labels=st.dictionaries(label_keys, texts, max_size=5),
    env=st.dictionaries(label_keys, texts, max_size=5),
    risk=st.lists(texts, max_size=4),
    tokens=st.integers(min_value=0, max_value=2**53),
    deadline=st.integers(min_value=0, max_value=2**53),
    prio=st.sampled_from(list(JobPriority)),
)
def test_jobrequest_codec_roundtrip(job_id, topic, tenant, labels, env, risk, tokens, deadline, prio):
    req = JobRequest(
        job_id=job_id, topic=topic, tenant_id=tenant, labels=labels, env=env,
        priority=prio, meta=JobMetadata(risk_tags=risk),
        budget=Budget(max_tokens=tokens, deadline_ms=deadline),
    )
    blob = req.encode()
    back = JobRequest.decode(blob)
    assert back.encode() == blob  # deterministic
    assert back.labels == labels and back.env == env
    assert back.meta.risk_tags == risk
    assert back.budget.max_tokens == tokens
    # json round trip preserves wire bytes
    assert JobRequest.from_dict(req.to_dict()).encode() == blob
    # hash is invariant to approval labels
    tampered = JobRequest.decode(blob)
    tampered.labels = dict(tampered.labels)
    tampered.labels["approval_granted"] = "true"
    assert job_hash(tampered) == job_hash(req)


@settings(max_examples=300, deadline=None)
@given(
    pattern=st.text(alphabet=st.sampled_from("ab*?[]-!/\\."), max_size=12),
    name=st.text(alphabet=st.sampled_from("ab./"), max_size=12),
)
def test_glob_match_never_crashes_and_agrees_with_fnmatch_subset(pattern, name):
    try:
        got = glob_match(pattern, name)
    except ValueError:
        return  # malformed pattern: Go returns ErrBadPattern; we raise
    assert isinstance(got, bool)
    # patterns without specials behave as equality
    if not any(c in pattern for c in "*?[]\\"):
        assert got == (pattern == name)
