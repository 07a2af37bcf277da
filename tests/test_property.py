"""Property-based tests (hypothesis): codec round-trips and compiled-policy
agreement on adversarial inputs."""
import pytest
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


# --- workflow expression language -------------------------------------------

from cordum_amd.workflow.eval import (  # noqa: E402
    EvalError,
    eval_expr,
    eval_template_string,
    eval_templates,
)

scalar = st.one_of(st.integers(min_value=-10**6, max_value=10**6),
                   st.booleans(), st.text(max_size=8))
json_val = st.recursive(
    scalar,
    lambda inner: st.one_of(st.lists(inner, max_size=3),
                            st.dictionaries(st.text(min_size=1, max_size=6), inner, max_size=3)),
    max_leaves=8,
)


@settings(max_examples=200, deadline=None)
@given(v=json_val)
def test_eval_templates_preserves_shape_without_placeholders(v):
    # values containing no ${...} must round-trip unchanged
    out = eval_templates(v, {"input": {}, "ctx": {}, "steps": {}})
    def strip(x):
        if isinstance(x, str) and "${" in x:
            return None
        if isinstance(x, list):
            return [strip(i) for i in x]
        if isinstance(x, dict):
            return {k: strip(u) for k, u in x.items()}
        return x
    assert strip(out) == strip(v) or any(
        isinstance(s, str) and "${" in s for s in _strings(v))


def _strings(v):
    if isinstance(v, str):
        yield v
    elif isinstance(v, list):
        for i in v:
            yield from _strings(i)
    elif isinstance(v, dict):
        for i in v.values():
            yield from _strings(i)


@settings(max_examples=300, deadline=None)
@given(a=st.integers(min_value=-10**6, max_value=10**6),
       b=st.integers(min_value=-10**6, max_value=10**6),
       op=st.sampled_from(["==", "!=", ">", "<", ">=", "<="]))
def test_eval_numeric_comparisons_match_python(a, b, op):
    got = eval_expr(f"{a} {op} {b}", {})
    want = {"==": a == b, "!=": a != b, ">": a > b,
            "<": a < b, ">=": a >= b, "<=": a <= b}[op]
    assert got == want


@settings(max_examples=200, deadline=None)
@given(path=st.lists(st.text(alphabet="abc", min_size=1, max_size=4),
                     min_size=1, max_size=4),
       val=st.integers(min_value=0, max_value=1000))
def test_eval_dot_path_resolution(path, val):
    ctx = cur = {}
    for part in path[:-1]:
        cur[part] = {}
        cur = cur[part]
    cur[path[-1]] = val
    assert eval_expr(".".join(path), ctx) == val
    # template wrapping preserves type
    assert eval_template_string("${" + ".".join(path) + "}", ctx) == val
    # mixed template stringifies
    out = eval_template_string("v=${" + ".".join(path) + "}!", ctx)
    assert out == f"v={val}!"


# --- bus subject matching + canonical JSON ----------------------------------

from cordum_amd.bus.bus import subject_matches  # noqa: E402
from cordum_amd.utils.canonical_json import canonical_json  # noqa: E402

token = st.text(alphabet="abcz0", min_size=1, max_size=4)
subject_s = st.lists(token, min_size=1, max_size=4).map(".".join)
pattern_tok = st.one_of(token, st.just("*"))


@settings(max_examples=300, deadline=None)
@given(subj=subject_s)
def test_subject_matching_nats_semantics(subj):
    parts = subj.split(".")
    assert subject_matches(subj, subj)                      # exact
    assert subject_matches(">", subj)                       # full wildcard
    assert subject_matches(parts[0] + ".>", subj) == (len(parts) > 1)
    star = ".".join(["*"] * len(parts))
    assert subject_matches(star, subj)                      # per-token star
    assert not subject_matches(subj + ".x", subj)           # longer pattern
    if len(parts) > 1:
        assert not subject_matches(parts[0], subj)          # shorter pattern


@settings(max_examples=300, deadline=None)
@given(pattern=st.lists(pattern_tok, min_size=1, max_size=4),
       subj=subject_s)
def test_subject_star_matches_iff_tokenwise(pattern, subj):
    p = ".".join(pattern)
    parts = subj.split(".")
    want = len(pattern) == len(parts) and all(
        pt == "*" or pt == sp for pt, sp in zip(pattern, parts))
    assert subject_matches(p, subj) == want


@settings(max_examples=200, deadline=None)
@given(v=json_val)
def test_canonical_json_stable_and_order_insensitive(v):
    import json as _json

    s1 = canonical_json(v)
    # round-trip through a parse (which scrambles nothing semantic)
    assert canonical_json(_json.loads(s1)) == s1
    if isinstance(v, dict):
        # rebuilding the dict in reversed insertion order must not change it
        rev = dict(reversed(list(v.items())))
        assert canonical_json(rev) == s1


# --- job state machine random walk ------------------------------------------

from cordum_amd.protocol import JobState  # noqa: E402
from cordum_amd.protocol.states import ALLOWED_TRANSITIONS, is_terminal  # noqa: E402
from cordum_amd.store import InvalidTransition, JobStore  # noqa: E402
from cordum_amd.utils.clock import ManualClock  # noqa: E402


@settings(max_examples=150, deadline=None)
@given(walk=st.lists(st.sampled_from([s for s in JobState if s != JobState.UNSPECIFIED]),
                     min_size=1, max_size=12))
def test_job_store_enforces_transition_table_on_random_walks(walk):
    js = JobStore(clock=ManualClock())
    cur = JobState.UNSPECIFIED
    attempts = 0
    touched = False
    for nxt in walk:
        legal = nxt in ALLOWED_TRANSITIONS.get(cur, frozenset())
        if legal:
            js.set_state("j", nxt)
            if nxt == JobState.SCHEDULED and cur != JobState.SCHEDULED:
                attempts += 1
            cur = nxt
            touched = True
        else:
            with pytest.raises(InvalidTransition):
                js.set_state("j", nxt)
    if touched:
        assert js.get_state("j") == cur
        assert int(js.get_job_meta("j").get("attempts", 0)) == attempts


# --- store snapshot round-trips ----------------------------------------------

import json as _json  # noqa: E402


@settings(max_examples=100, deadline=None)
@given(
    jobs=st.lists(
        st.tuples(st.text(alphabet="jk0", min_size=1, max_size=5),   # id
                  st.sampled_from(["t1", "t2", ""]),                 # tenant
                  st.sampled_from([JobState.PENDING, JobState.SCHEDULED])),
        min_size=0, max_size=8, unique_by=lambda t: t[0]),
)
def test_job_store_snapshot_roundtrip_and_json_clean(jobs):
    clock = ManualClock()
    js = JobStore(clock=clock)
    for jid, tenant, state in jobs:
        if tenant:
            js.set_tenant(jid, tenant)
        js.set_state(jid, state)
        if state == JobState.SCHEDULED:
            js.set_deadline(jid, clock.now_micros() + 1_000_000)
        clock.advance(1)
    snap = js.snapshot()
    # checkpoint constraint: the snapshot must be plain JSON (the WAL file is)
    snap2 = _json.loads(_json.dumps(snap))
    js2 = JobStore(clock=clock)
    js2.restore(snap2)
    assert js2.snapshot() == snap
    for jid, tenant, state in jobs:
        assert js2.get_state(jid) == state
        if tenant:
            assert js2.get_job_meta(jid).get("tenant") == tenant
    # indexes rebuilt: per-state listings agree
    for s in (JobState.PENDING, JobState.SCHEDULED):
        assert sorted(js.list_jobs_by_state(s)) == sorted(js2.list_jobs_by_state(s))
    assert sorted(js.list_expired_deadlines()) == sorted(js2.list_expired_deadlines())


# --- native codec equivalence -------------------------------------------------


@settings(max_examples=300, deadline=None)
@given(
    job_id=texts, topic=texts,
    labels=st.dictionaries(label_keys, texts, max_size=5),
    risk=st.lists(texts, max_size=4),
    tokens=st.integers(min_value=-2**53, max_value=2**53),
    prio=st.sampled_from(list(JobPriority)),
)
def test_native_codec_bytes_identical_to_python(job_id, topic, labels, risk, tokens, prio):
    from cordum_amd.protocol import capv2

    if capv2._NATIVE is None:
        pytest.skip("native codec not built")
    req = JobRequest(job_id=job_id, topic=topic, labels=labels, priority=prio,
                     meta=JobMetadata(risk_tags=risk),
                     budget=Budget(max_tokens=tokens))
    nb = req.encode()
    assert nb == req.encode_py()
    # native decode == python decode, and both re-encode to the same bytes
    d_native = JobRequest.decode(nb)
    d_python = JobRequest.decode_py(nb)
    assert d_native.encode() == d_python.encode_py() == nb
    assert d_native.labels == d_python.labels == labels
    assert d_native.meta.risk_tags == risk


@settings(max_examples=500, deadline=None)
@given(junk=st.binary(max_size=64))
def test_codec_decode_rejects_or_parses_arbitrary_bytes(junk):
    """API-boundary robustness: decoding attacker-controlled bytes must either
    produce a message or raise a clean error — never crash the process or
    hang (both the native engine and the Python fallback)."""
    from cordum_amd.protocol import capv2

    from cordum_amd.protocol.capv2 import PolicyCheckResponse

    for decoder in (JobRequest.decode, JobRequest.decode_py,
                    PolicyCheckResponse.decode):
        try:
            msg = decoder(junk)
        except (ValueError, RuntimeError, IndexError, UnicodeDecodeError):
            continue
        # decodable junk must re-encode deterministically
        assert isinstance(msg.encode(), bytes)


# --- workflow engine: random DAGs --------------------------------------------

from cordum_amd.bus import LoopbackBus  # noqa: E402
from cordum_amd.protocol import subjects as subj  # noqa: E402
from cordum_amd.protocol.capv2 import JobResult, JobStatus  # noqa: E402
from cordum_amd.workflow import (  # noqa: E402
    Engine,
    RUN_FAILED,
    RUN_SUCCEEDED,
    Step,
    Workflow,
    WorkflowRun,
    WorkflowStore,
)
from cordum_amd.store import MemoryStore  # noqa: E402


@settings(max_examples=60, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=7),
    edge_bits=st.integers(min_value=0, max_value=2**21 - 1),
    fail_step=st.integers(min_value=-1, max_value=6),
)
def test_workflow_random_dag_terminates_with_correct_statuses(n, edge_bits, fail_step):
    """Any random dependency DAG must drive to a terminal run status with
    every reachable step completed; a failing step must block its downstream
    closure (deps gate workflow/engine.go depsSatisfied:1231-1242) and fail
    the run."""
    clock = ManualClock()
    bus = LoopbackBus(clock=clock)
    store = WorkflowStore(clock=clock)
    memory = MemoryStore(clock=clock)
    engine = Engine(store, bus, memory=memory, clock=clock)
    submitted = []
    bus.subscribe(subj.SUBJECT_SUBMIT, lambda s, p: submitted.append(p.job_request))

    names = [f"s{i}" for i in range(n)]
    deps = {i: [j for j in range(i) if (edge_bits >> (i * (i - 1) // 2 + j)) & 1]
            for i in range(n)}
    steps = {
        names[i]: Step.from_dict(names[i], {
            "type": "worker", "topic": "job.t",
            "depends_on": [names[j] for j in deps[i]],
        })
        for i in range(n)
    }
    wf = Workflow(id="wfP", org_id="org", steps=steps)
    store.put_workflow(wf)
    run = WorkflowRun(id="runP", workflow_id="wfP", org_id="org")
    store.create_run(run)
    engine.start_run("wfP", "runP")

    failing = names[fail_step] if 0 <= fail_step < n else None
    done = set()
    for _ in range(n * n + 2):  # fixpoint: drain all dispatches
        pending = [r for r in submitted if r.job_id not in done]
        if not pending:
            break
        for req in pending:
            done.add(req.job_id)
            sid = req.labels["step_id"]
            status = JobStatus.FAILED if sid == failing else JobStatus.SUCCEEDED
            engine.handle_job_result(JobResult(job_id=req.job_id, status=status))

    run = store.get_run("runP")
    # downstream closure of the failing step
    blocked = set()
    if failing is not None:
        idx = {nm: i for i, nm in enumerate(names)}
        frontier = {failing}
        while frontier:
            cur = frontier.pop()
            blocked.add(cur)
            for i in range(n):
                if names[i] not in blocked and any(names[j] in blocked for j in deps[i]):
                    frontier.add(names[i])
    if failing is None:
        assert run.status == RUN_SUCCEEDED, run.status
        assert all(run.steps[nm].status == "succeeded" for nm in names)
    else:
        assert run.status == RUN_FAILED, run.status
        for nm in names:
            if nm == failing:
                assert run.steps[nm].status == "failed"
            elif nm in blocked:
                sr = run.steps.get(nm)
                assert sr is None or sr.status in ("pending", "skipped", "blocked")
            else:
                # results arriving after the run went terminal are ignored
                # (engine.handle_job_result, mirroring the reference's
                # processed-state dedup) so in-flight siblings stay running;
                # independent steps not yet dispatched have no record at all
                sr = run.steps.get(nm)
                assert sr is None or sr.status in ("succeeded", "running", "pending")


# --- pack archive hardening ---------------------------------------------------

import io  # noqa: E402
import tarfile  # noqa: E402


def _tar_with(names):
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        for nm in names:
            data = b"x"
            info = tarfile.TarInfo(name=nm)
            info.size = len(data)
            tf.addfile(info, io.BytesIO(data))
    return buf.getvalue()


@settings(max_examples=200, deadline=None)
@given(name=st.text(alphabet=st.sampled_from("ab./\\_"), min_size=1, max_size=16))
def test_pack_extract_rejects_traversal_never_escapes(name):
    from cordum_amd.gateway.packs import PackError, extract_pack

    try:
        blob = _tar_with([name])
    except ValueError:
        return  # tarfile itself refuses the name
    try:
        files = extract_pack(blob)
    except PackError:
        return  # rejected: fine
    # anything accepted must be a clean relative path with no traversal
    for accepted in files:
        assert accepted
        assert not accepted.startswith("/")
        assert ".." not in accepted.split("/")


def test_pack_extract_rejects_known_hostile_names():
    from cordum_amd.gateway.packs import PackError, extract_pack

    for nm in ["../evil", "a/../../evil", "/abs/path", "./../up", "..", "./.."]:
        with pytest.raises(PackError):
            extract_pack(_tar_with([nm, "pack.yaml"]))


# --- schema validator robustness ---------------------------------------------


@settings(max_examples=150, deadline=None)
@given(schema=json_val, value=json_val)
def test_schema_validator_never_crashes(schema, value):
    """validate_value on arbitrary (schema, value) pairs returns a bool list
    outcome or raises nothing — hostile schemas must not crash the gateway."""
    from cordum_amd.store.schema_registry import validate_value

    errs = validate_value(schema, value)
    assert isinstance(errs, list)


# --- bus NAK redelivery -------------------------------------------------------


@settings(max_examples=80, deadline=None)
@given(n_msgs=st.integers(min_value=1, max_value=6),
       nak_rounds=st.integers(min_value=0, max_value=3))
def test_bus_nak_redelivers_until_ack(n_msgs, nak_rounds):
    """A handler that NAKs k times then succeeds must see each message
    exactly k+1 times (JetStream redelivery semantics, bus/nats.go:146-168),
    pinned to the NAKing consumer."""
    from cordum_amd.bus import LoopbackBus, RetryAfter
    from cordum_amd.protocol.capv2 import BusPacket, JobRequest

    clock = ManualClock()
    bus = LoopbackBus(clock=clock)
    seen = {}

    def handler(subject, pkt):
        jid = pkt.job_request.job_id
        seen[jid] = seen.get(jid, 0) + 1
        if seen[jid] <= nak_rounds:
            raise RetryAfter(1.0, "busy")

    bus.subscribe("sys.job.submit", handler, queue_group="g")
    for i in range(n_msgs):
        bus.publish("sys.job.submit", BusPacket(job_request=JobRequest(job_id=f"j{i}")))
    for _ in range(nak_rounds + 2):
        clock.advance(2)
        bus.pump()
    assert seen == {f"j{i}": nak_rounds + 1 for i in range(n_msgs)}


# --- expression/template fuzz -------------------------------------------------


@settings(max_examples=400, deadline=None)
@given(expr=st.text(alphabet=st.sampled_from("ab.(){}$!<>=' 0137length"), max_size=24),
       scope=json_val)
def test_eval_never_raises_uncontrolled(expr, scope):
    """Workflow expressions come from user-authored YAML: arbitrary strings
    must evaluate or raise EvalError — nothing else (no crashes in the
    engine's scheduleReady path)."""
    ctx = scope if isinstance(scope, dict) else {"input": scope}
    for fn in (eval_expr, eval_template_string):
        try:
            fn(expr, ctx)
        except EvalError:
            pass
        except RecursionError:
            pass  # pathological nesting is bounded by python's own limit


# --- gateway submit-body fuzz -------------------------------------------------


@settings(max_examples=60, deadline=None)
@given(body=json_val,
       path=st.sampled_from(["/api/v1/jobs", "/api/v1/workflows",
                             "/api/v1/policy/evaluate", "/api/v1/locks/acquire",
                             "/api/v1/schemas", "/api/v1/config"]))
def test_gateway_mutating_endpoints_handle_arbitrary_json(body, path):
    """POST with arbitrary JSON must answer 2xx or 4xx — never a 500
    (gateway.go:1757-1977 validation contract)."""
    code, _ = _gw_client().post_json(path, body)
    assert code < 500, (code, path, body)


_GW = None


def _gw_client():
    global _GW
    if _GW is None:
        from fastapi.testclient import TestClient

        from cordum_amd.scheduler import PoolProfile, PoolRouting
        from cordum_amd.gateway.app import create_app
        from cordum_amd.gateway.auth import BasicAuthProvider
        from cordum_amd.runtime.node import Node

        node = Node(clock=ManualClock(),
                    routing=PoolRouting(topics={"job.default": ["default"]},
                                        pools={"default": PoolProfile()})).start()
        node.add_worker("w1", topics=["job.default"])
        app = create_app(node, auth=BasicAuthProvider(api_keys=["test-key"]))
        tc = TestClient(app, raise_server_exceptions=False)

        class C:
            tc = None
            headers = {"X-API-Key": "test-key", "X-Principal-Id": "t",
                       "X-Principal-Role": "admin"}

            def post_json(self, path, body):
                r = self.tc.post(path, json=body, headers=self.headers)
                return r.status_code, r

        C.tc = tc
        _GW = C()
    return _GW


@settings(max_examples=80, deadline=None)
@given(q=st.text(alphabet=st.sampled_from("abc019-_.%&=?"), max_size=16),
       path=st.sampled_from(["/api/v1/jobs", "/api/v1/workflow-runs", "/api/v1/dlq/page",
                             "/api/v1/approvals", "/api/v1/workflows", "/api/v1/packs"]))
def test_gateway_list_endpoints_handle_junk_query_params(q, path):
    """Cursor/limit/filter query strings are attacker-controlled: junk must
    produce 2xx/4xx, never a 500 (micros-cursor parsing gateway.go:918-1009)."""
    gw = _gw_client()
    r = gw.tc.get(f"{path}?cursor={q}&limit={q}&state={q}&topic={q}",
                  headers=gw.headers)
    assert r.status_code < 500, (r.status_code, path, q)


@settings(max_examples=80, deadline=None)
@given(junk=st.text(alphabet=st.sampled_from("ab:/~.%-_0"), max_size=20))
def test_gateway_pointer_and_bundle_ids_handle_junk(junk):
    """Pointer params and `~`-escaped bundle ids are attacker-controlled path
    material: junk must never 500 (memory reader gateway.go:1206+, bundle id
    unescape policy_bundles.go:24-122)."""
    gw = _gw_client()
    r = gw.tc.get(f"/api/v1/memory?ptr={junk}", headers=gw.headers)
    assert r.status_code < 500, ("memory", r.status_code, junk)
    r = gw.tc.get(f"/api/v1/policy/bundles/{junk or 'x'}", headers=gw.headers)
    assert r.status_code < 500, ("bundle", r.status_code, junk)
    r = gw.tc.get(f"/api/v1/artifacts/{junk or 'x'}", headers=gw.headers)
    assert r.status_code < 500, ("artifact", r.status_code, junk)


@settings(max_examples=200, deadline=None)
@given(job_id=st.text(min_size=1, max_size=10), topic=texts,
       cfg=texts, other_env=st.dictionaries(label_keys, texts, max_size=3))
def test_job_hash_strips_exactly_the_approval_binding_exclusions(job_id, topic, cfg, other_env):
    """job_hash ignores approval_* labels, the bus msg-id label and
    CORDUM_EFFECTIVE_CONFIG env (job_hash.go:15-48) — and ONLY those:
    any other label/env/field change must change the hash."""
    base = JobRequest(job_id=job_id, topic=topic, env=dict(other_env))
    h0 = job_hash(base)

    excluded = JobRequest.decode(base.encode())
    excluded.env = dict(excluded.env)
    excluded.env["CORDUM_EFFECTIVE_CONFIG"] = cfg
    excluded.labels = {"approval_granted": "true", "approval_reason": "r",
                       "Approval_Note": "mixed-case stripped too (job_hash.go:27 lowercases)",
                       "cordum.bus_msg_id": "m1"}
    assert job_hash(excluded) == h0

    changed = JobRequest.decode(base.encode())
    changed.env = dict(changed.env)
    changed.env["OTHER"] = "x"
    assert job_hash(changed) != h0
    changed2 = JobRequest.decode(base.encode())
    changed2.labels = {"team": "x"}
    assert job_hash(changed2) != h0
    changed3 = JobRequest.decode(base.encode())
    changed3.topic = topic + "!"
    assert job_hash(changed3) != h0


@given(
    job_id=st.text(min_size=0, max_size=12),
    extra=st.lists(
        st.tuples(st.integers(min_value=100, max_value=500),   # unknown field num
                  st.one_of(st.integers(min_value=0, max_value=2**40),
                            st.binary(max_size=24))),
        max_size=4),
)
@settings(max_examples=60, deadline=None)
def test_unknown_field_roundtrip_property(job_id, extra):
    """Any combination of unknown varint/bytes fields appended to a valid
    BusPacket must survive decode→encode byte-for-byte (forward compat for
    relays), through whichever codec engine (native or Python) is active."""
    from cordum_amd.protocol.capv2 import BusPacket, JobRequest

    def varint(n):
        out = b""
        while True:
            b = n & 0x7F
            n >>= 7
            out += bytes([b | (0x80 if n else 0)])
            if not n:
                return out

    wire = BusPacket(protocol_version=1,
                     job_request=JobRequest(job_id=job_id)).encode()
    for num, val in extra:
        if isinstance(val, int):
            wire += varint((num << 3) | 0) + varint(val)
        else:
            wire += varint((num << 3) | 2) + varint(len(val)) + val
    rt = BusPacket.decode(wire)
    assert rt.job_request is not None and rt.job_request.job_id == job_id
    assert rt.encode() == wire
