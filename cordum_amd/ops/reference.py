"""Torch reference implementations of every HIP kernel.

These are the numerics oracles for the GPU tests (plain torch, fp32/int
exact) and the CPU execution path for environments without a GPU. Each
function mirrors its kernel in cordum_amd/ops/hip/cordum_kernels.hip
bit-for-bit (deterministic tie-breaks included).
"""
from __future__ import annotations


import torch

from ..protocol.states import transition_lut

OVERLOAD = 0.9


def least_loaded_pick_ref(
    w_pool: torch.Tensor,
    w_active: torch.Tensor,
    w_maxp: torch.Tensor,
    w_cpu: torch.Tensor,
    w_gpu: torch.Tensor,
    w_labels: torch.Tensor,
    j_poolmask: torch.Tensor,
    j_labels: torch.Tensor,
) -> torch.Tensor:
    """K2 oracle. Returns int32 [J]: worker idx, -1 no_workers, -2 overloaded."""
    NW = w_pool.shape[0]
    NJ = j_poolmask.shape[0]
    if NJ == 0:
        return torch.empty(0, dtype=torch.int32)
    pool_ok = ((j_poolmask.unsqueeze(1) >> w_pool.clamp(0, 63).unsqueeze(0)) & 1).bool()
    pool_ok &= (w_pool >= 0).unsqueeze(0) & (w_pool < 64).unsqueeze(0)
    labels_ok = (j_labels.unsqueeze(1) & ~w_labels.unsqueeze(0)) == 0
    eligible = pool_ok & labels_ok  # [J, W]

    util_over = (w_maxp > 0) & (w_active.float() / w_maxp.clamp(min=1).float() >= OVERLOAD)
    over = util_over | (w_cpu >= 90) | (w_gpu >= 90)  # [W]
    usable = eligible & ~over.unsqueeze(0)

    score = w_active.float() + w_cpu * 0.01 + w_gpu * 0.01
    key = (score.view(torch.int32).to(torch.int64) << 32) | torch.arange(NW, dtype=torch.int64)
    big = torch.iinfo(torch.int64).max
    keys = torch.where(usable, key.unsqueeze(0).expand(NJ, NW), torch.full((NJ, NW), big, dtype=torch.int64))
    best = keys.min(dim=1).values
    pick = (best & 0xFFFFFFFF).to(torch.int32)

    n_eligible = eligible.sum(dim=1)
    n_over = (eligible & over.unsqueeze(0)).sum(dim=1)
    none = best == big
    all_over = none & (n_eligible > 0) & (n_over == n_eligible)
    out = torch.where(none, torch.full_like(pick, -1), pick)
    out = torch.where(all_over, torch.full_like(pick, -2), out)
    return out


def worker_precompute_ref(w_pool, w_active, w_maxp, w_cpu, w_gpu) -> torch.Tensor:
    """K2a oracle: per-worker (score, idx) key; overloaded -> OVERLOADED tag."""
    NW = w_pool.shape[0]
    idx = torch.arange(NW, dtype=torch.int64)
    util_over = (w_maxp > 0) & (w_active.float() / w_maxp.clamp(min=1).float() >= OVERLOAD)
    over = util_over | (w_cpu >= 90) | (w_gpu >= 90)
    score = w_active.float() + w_cpu * 0.01 + w_gpu * 0.01
    key = (score.view(torch.int32).to(torch.int64) << 32) | idx
    # 0xFFFFFFFE00000000 as signed int64 (torch has no uint64)
    over_key = torch.tensor(0xFFFFFFFE00000000 - (1 << 64), dtype=torch.int64) | idx
    return torch.where(over, over_key, key)


def least_loaded_pick_keys_ref(
    w_pool: torch.Tensor,
    w_keys: torch.Tensor,
    w_labels: torch.Tensor,
    j_poolmask: torch.Tensor,
    j_labels: torch.Tensor,
) -> torch.Tensor:
    """K2 oracle over precomputed keys (must agree with least_loaded_pick_ref)."""
    NW = w_pool.shape[0]
    NJ = j_poolmask.shape[0]
    if NJ == 0:
        return torch.empty(0, dtype=torch.int32)
    pool_ok = ((j_poolmask.unsqueeze(1) >> w_pool.clamp(0, 63).unsqueeze(0)) & 1).bool()
    pool_ok &= (w_pool >= 0).unsqueeze(0) & (w_pool < 64).unsqueeze(0)
    labels_ok = (j_labels.unsqueeze(1) & ~w_labels.unsqueeze(0)) == 0
    eligible = pool_ok & labels_ok
    over = ((w_keys >> 32) & 0xFFFFFFFF) == 0xFFFFFFFE
    usable = eligible & ~over.unsqueeze(0)
    big = torch.iinfo(torch.int64).max
    keys = torch.where(usable, w_keys.unsqueeze(0).expand(NJ, NW), torch.full((NJ, NW), big, dtype=torch.int64))
    best = keys.min(dim=1).values
    pick = (best & 0xFFFFFFFF).to(torch.int32)
    n_eligible = eligible.sum(dim=1)
    n_over = (eligible & over.unsqueeze(0)).sum(dim=1)
    none = best == big
    all_over = none & (n_eligible > 0) & (n_over == n_eligible)
    out = torch.where(none, torch.full_like(pick, -1), pick)
    out = torch.where(all_over, torch.full_like(pick, -2), out)
    return out


def echo_execute_indexed_ref(ctx_arena, slots, res_arena, res_sum, stride: int):
    N = ctx_arena.numel() // stride
    ctx = ctx_arena.view(N, stride)
    res = res_arena.view(N, stride)
    sl = slots.long()
    res[sl] = ctx[sl]
    sums = ctx[sl].to(torch.int64).sum(dim=1).remainder(1 << 32).to(torch.int32)
    res_sum[sl] = sums
    return res_sum


def apply_transitions_ref(
    states: torch.Tensor,
    attempts: torch.Tensor,
    deadlines: torch.Tensor,
    slots: torch.Tensor,
    to_states: torch.Tensor,
) -> torch.Tensor:
    """K5 oracle (sequential; duplicate slots in one batch are applied in
    order, matching the kernel's per-slot independence assumption — the
    pipeline never puts the same slot twice in one batch)."""
    lut = torch.tensor(transition_lut(), dtype=torch.uint8)
    ok = torch.zeros(slots.shape[0], dtype=torch.uint8)
    for i in range(slots.shape[0]):
        s = int(slots[i])
        frm = int(states[s])
        to = int(to_states[i])
        if lut[frm, to]:
            ok[i] = 1
            states[s] = to
            if to == 3 and frm != 3:
                attempts[s] += 1
            if to >= 6:
                deadlines[s] = torch.iinfo(torch.int64).max
    return ok


def deadline_scan_ref(
    states: torch.Tensor,
    deadlines: torch.Tensor,
    updated_at: torch.Tensor,
    now_us: int,
    dispatch_cutoff_us: int,
    running_cutoff_us: int,
) -> torch.Tensor:
    """K4 oracle: sorted slot indices due for TIMEOUT."""
    active = (states >= 1) & (states <= 5)
    expired = active & (deadlines <= now_us)
    expired |= (states == 4) & (updated_at <= dispatch_cutoff_us)
    expired |= (states == 5) & (updated_at <= running_cutoff_us)
    return torch.nonzero(expired, as_tuple=False).flatten().to(torch.int32)


def echo_execute_ref(ctx_arena: torch.Tensor, res_arena: torch.Tensor, stride: int) -> torch.Tensor:
    """Echo worker oracle: copy payload, return per-job uint32 checksum."""
    B = ctx_arena.numel() // stride
    ctx = ctx_arena.view(B, stride)
    res_arena.view(B, stride).copy_(ctx)
    return ctx.to(torch.int64).sum(dim=1).remainder(1 << 32).to(torch.int32)


# device run-table step-state encoding (K3): 0 pending, 1 running, 2 waiting,
# 3 succeeded, 4 failed, 5 cancelled, 6 timed_out
RUN_STEP_PENDING, RUN_STEP_SUCCEEDED = 0, 3


def run_readiness_ref(step_state, deps_mask, n_steps, run_active):
    """K3 oracle: per-run 64-bit ready mask + (run, step) dispatch pairs."""
    NR = step_state.shape[0]
    ready = torch.zeros(NR, dtype=torch.int64)
    pairs = []
    for r in range(NR):
        if not bool(run_active[r]):
            continue
        ns = int(n_steps[r])
        succeeded = 0
        for t in range(ns):
            if int(step_state[r, t]) == RUN_STEP_SUCCEEDED:
                succeeded |= 1 << t
        m = 0
        for s_i in range(ns):
            if int(step_state[r, s_i]) != RUN_STEP_PENDING:
                continue
            need = int(deps_mask[r, s_i]) & ((1 << 64) - 1)
            if need & ~succeeded & ((1 << 64) - 1):
                continue
            m |= 1 << s_i
            pairs.append((r, s_i))
        ready[r] = m - (1 << 64) if m >= (1 << 63) else m
    return ready, pairs


# -- padded cross-rank dispatch references ------------------------------------


RQ_MAX_DELIVER = 5


def _rq_dead_mark(rq_dead, dead_src, dead_count, src: int) -> None:
    rq_dead[0] += 1
    if dead_src is not None:
        dp = int(dead_count[0])
        dead_count[0] = dp + 1
        if dp < dead_src.shape[0]:
            dead_src[dp] = src


def _rq_park(rq_src, rq_widx, rq_attempts, rq_count, rq_dead,
             src: int, widx: int, attempts: int,
             dead_src=None, dead_count=None) -> None:
    p = int(rq_count[0])
    rq_count[0] = p + 1
    if p < rq_src.shape[0]:
        rq_src[p] = src
        rq_widx[p] = widx
        rq_attempts[p] = attempts
    else:
        _rq_dead_mark(rq_dead, dead_src, dead_count, src)


def pack_by_dest_ref(routable_slots, routable_widx, routable_count,
                     send_slots, send_widx, send_cnt, nwl: int, cap: int,
                     rq_src, rq_widx, rq_attempts, rq_count, rq_dead,
                     dead_src=None, dead_count=None):
    n = int(routable_count[0])
    for i in range(n):
        slot = int(routable_slots[i])
        widx = int(routable_widx[i])
        dest = widx // nwl
        pos = int(send_cnt[dest])
        send_cnt[dest] += 1
        if pos < cap:
            send_slots[dest * cap + pos] = slot
            send_widx[dest * cap + pos] = widx % nwl
        else:
            _rq_park(rq_src, rq_widx, rq_attempts, rq_count, rq_dead,
                     slot, widx, 1, dead_src, dead_count)


def pack_requeue_ref(rq_prev_widx, rq_prev_attempts, rq_prev_count,
                     send_slots, send_widx, send_cnt, nwl: int, cap: int,
                     rq_src, rq_widx, rq_attempts, rq_count, rq_dead,
                     dead_src=None, dead_count=None):
    n = min(int(rq_prev_count[0]), int(rq_prev_widx.shape[0]))
    for j in range(n):
        widx = int(rq_prev_widx[j])
        dest = widx // nwl
        pos = int(send_cnt[dest])
        send_cnt[dest] += 1
        if pos < cap:
            send_slots[dest * cap + pos] = -1 - j
            send_widx[dest * cap + pos] = widx % nwl
        else:
            att = int(rq_prev_attempts[j]) + 1
            if att > RQ_MAX_DELIVER:
                _rq_dead_mark(rq_dead, dead_src, dead_count, -1 - j)
            else:
                _rq_park(rq_src, rq_widx, rq_attempts, rq_count, rq_dead,
                         -1 - j, widx, att, dead_src, dead_count)


def materialize_rq_payload_ref(payload, rq_prev_payload, rq_src, rq_count,
                               rq_payload, stride: int):
    pl = payload.view(-1, stride)
    prev = rq_prev_payload.view(-1, stride)
    dst = rq_payload.view(-1, stride)
    n = min(int(rq_count[0]), int(rq_src.shape[0]))
    for j in range(n):
        s = int(rq_src[j])
        dst[j] = pl[s] if s >= 0 else prev[-1 - s]


def gather_payload_padded_ref(payload, rq_prev_payload, send_slots, send_cnt,
                              send_payload, stride: int, cap: int, world: int):
    pl = payload.view(-1, stride)
    prev = rq_prev_payload.view(-1, stride)
    sp = send_payload.view(-1, stride)
    for r in range(world):
        for e in range(int(send_cnt[r])):
            i = r * cap + e
            s = int(send_slots[i])
            sp[i] = pl[s] if s >= 0 else prev[-1 - s]


def echo_padded_ref(recv_payload, recv_cnt, res_arena, res_sums,
                    stride: int, cap: int, world: int):
    rp = recv_payload.view(-1, stride)
    ra = res_arena.view(-1, stride)
    for r in range(world):
        for e in range(int(recv_cnt[r])):
            i = r * cap + e
            ra[i] = rp[i]
            res_sums[i] = int(rp[i].to(torch.int64).sum()) % (1 << 32) - (
                (1 << 32) if int(rp[i].to(torch.int64).sum()) % (1 << 32) >= (1 << 31) else 0
            )


def apply_transitions_padded_ref(states, attempts, deadlines, slots_pad, cnt,
                                 to_state: int, cap: int, world: int):
    from ..protocol.states import transition_lut

    lut = transition_lut()
    for r in range(world):
        for e in range(int(cnt[r])):
            slot = int(slots_pad[r * cap + e])
            if slot < 0:  # redelivered entry: its batch slot was recycled
                continue
            frm = int(states[slot])
            if lut[frm][to_state]:
                states[slot] = to_state
                if to_state == 3 and frm != 3:
                    attempts[slot] += 1
                if to_state >= 6:
                    deadlines[slot] = torch.iinfo(torch.int64).max


def load_feedback_padded_ref(recv_widx, recv_cnt, w_active_local, cap: int, world: int):
    for r in range(world):
        for e in range(int(recv_cnt[r])):
            w_active_local[int(recv_widx[r * cap + e])] += 1


# -- K3-WF device workflow engine references ----------------------------------
# States/kinds mirror the kernel constants (cordum_kernels.hip K3-WF section).

WFS_PENDING, WFS_DISPATCHED, WFS_WAITING = 0, 1, 2
WFS_SUCCEEDED, WFS_FAILED, WFS_SKIPPED = 3, 4, 5
WFK_WORKER, WFK_FOR_EACH, WFK_APPROVAL, WFK_CONDITION, WFK_DELAY = 0, 1, 2, 3, 4


def wf_sweep_ref(step_state, deps_mask, n_steps, run_active, step_kind,
                 cond_bits, next_ready, tick,
                 disp_runs, disp_steps, disp_count,
                 appr_runs, appr_steps, appr_count):
    tick = int(tick[0]) if hasattr(tick, "__getitem__") else int(tick)
    NR = int(n_steps.shape[0])
    cap, acap = int(disp_runs.shape[0]), int(appr_runs.shape[0])
    for run in range(NR):
        if not int(run_active[run]):
            continue
        ns = int(n_steps[run])
        base = run * 64
        satisfied = 0
        for t in range(ns):
            if int(step_state[base + t]) in (WFS_SUCCEEDED, WFS_SKIPPED):
                satisfied |= 1 << t
        for s in range(ns):
            if int(step_state[base + s]) != WFS_PENDING:
                continue
            need = int(deps_mask[base + s]) & ((1 << 64) - 1)
            if need & ~satisfied & ((1 << 64) - 1):
                continue
            if int(next_ready[base + s]) > tick:
                continue
            kind = int(step_kind[base + s])
            if kind in (WFK_WORKER, WFK_FOR_EACH):
                pos = int(disp_count[0])
                disp_count[0] = pos + 1
                if pos < cap:
                    disp_runs[pos] = run
                    disp_steps[pos] = s
            elif kind == WFK_APPROVAL:
                step_state[base + s] = WFS_WAITING
                pos = int(appr_count[0])
                appr_count[0] = pos + 1
                if pos < acap:
                    appr_runs[pos] = run
                    appr_steps[pos] = s
            elif kind == WFK_CONDITION:
                step_state[base + s] = (
                    WFS_SUCCEEDED if (int(cond_bits[run]) >> s) & 1 else WFS_SKIPPED)
            elif kind == WFK_DELAY:
                step_state[base + s] = WFS_SUCCEEDED


def wf_expand_ref(disp_runs, disp_steps, disp_count, step_state,
                  children_todo, children_out, max_parallel, child_tag,
                  child_seq, child_widx, child_count, children_emitted,
                  dispatch_tick, tick, order, valid_count):
    tick = int(tick[0]) if hasattr(tick, "__getitem__") else int(tick)
    CB = int(child_tag.shape[0])
    n = min(int(disp_count[0]), int(disp_runs.shape[0]))
    V = min(max(1, int(valid_count[0])), 1024)
    for e in range(n):
        run, step = int(disp_runs[e]), int(disp_steps[e])
        rs = run * 64 + step
        todo = int(children_todo[rs])
        # for_each max_parallel windowing (dataflow_test.go:71 semantics)
        maxp = int(max_parallel[rs])
        if maxp > 0:
            room = maxp - int(children_out[rs])
            if room < todo:
                todo = max(0, room)
        # monotone per-step emission counter -> deterministic per-child
        # ordinal stream (matches the kernel; retries get fresh ordinals)
        seq0 = int(children_emitted[rs])
        if todo <= 0:
            if int(children_out[rs]) > 0:
                step_state[rs] = WFS_DISPATCHED
            continue
        base = int(child_count[0])
        if base + todo > CB:
            continue  # arena full: retry next sweep
        child_count[0] = base + todo
        step_state[rs] = WFS_DISPATCHED
        children_out[rs] += todo
        children_emitted[rs] += todo
        children_todo[rs] -= todo
        dispatch_tick[rs] = tick
        tag = run * 64 + step
        for k in range(todo):
            child_tag[base + k] = tag
            child_seq[base + k] = seq0 + k
            child_widx[base + k] = int(order[(base + k) % V])


def _wf_mix(x: int) -> int:
    x &= 0xFFFFFFFF
    x ^= x >> 16
    x = (x * 0x7FEB352D) & 0xFFFFFFFF
    x ^= x >> 15
    x = (x * 0x846CA68B) & 0xFFFFFFFF
    x ^= x >> 16
    return x


def wf_fail_hash(tag: int, seq: int) -> int:
    return _wf_mix((tag * 2654435761 ^ seq * 40503) & 0xFFFFFFFF)


def wf_drop_hash(tag: int, seq: int) -> int:
    return _wf_mix((tag * 0x85EBCA6B ^ seq * 0xC2B2AE35) & 0xFFFFFFFF)


def wf_apply_ref(send_slots, send_cnt, child_tag, child_seq, rq_prev_tag,
                 rq_prev_seq, children_done, children_fail, children_out,
                 fail_ppt: int, drop_ppt: int, cap: int, world: int):
    for r in range(world):
        for e in range(min(int(send_cnt[r]), cap)):
            i = r * cap + e
            s = int(send_slots[i])
            tag = int(child_tag[s]) if s >= 0 else int(rq_prev_tag[-1 - s])
            seq = int(child_seq[s]) if s >= 0 else int(rq_prev_seq[-1 - s])
            if wf_drop_hash(tag, seq) % 1000 < drop_ppt:
                continue  # lost result: K4-WF timeout scan recovers it
            if wf_fail_hash(tag, seq) % 1000 < fail_ppt:
                children_fail[tag] += 1
            else:
                children_done[tag] += 1
            children_out[tag] -= 1


def wf_apply_dead_ref(dead_src, dead_count, child_tag, rq_prev_tag,
                      children_fail, children_out):
    n = min(int(dead_count[0]), int(dead_src.shape[0]))
    for i in range(n):
        s = int(dead_src[i])
        tag = int(child_tag[s]) if s >= 0 else int(rq_prev_tag[-1 - s])
        children_fail[tag] += 1
        children_out[tag] -= 1


def wf_commit_ref(step_state, step_attempts, children_todo, children_out,
                  children_done, children_fail, max_parallel, next_ready,
                  tick, max_retries: int, retry_count=None):
    tick = int(tick[0]) if hasattr(tick, "__getitem__") else int(tick)
    for i in range(int(step_state.numel())):
        if int(step_state[i]) != WFS_DISPATCHED:
            continue
        if int(children_out[i]) != 0:
            # window slide / arena resume: more work and room -> re-open
            if int(children_todo[i]) > 0 and (
                int(max_parallel[i]) == 0
                or int(children_out[i]) < int(max_parallel[i])
            ):
                step_state[i] = WFS_PENDING
            continue
        fail = int(children_fail[i])
        if fail > 0:
            if int(step_attempts[i]) < max_retries:
                step_attempts[i] += 1
                children_todo[i] += fail
                children_fail[i] = 0
                next_ready[i] = tick + min(1 << int(step_attempts[i]), 16)
                step_state[i] = WFS_PENDING
                if retry_count is not None:
                    retry_count[0] += fail
            else:
                step_state[i] = WFS_FAILED
        elif int(children_todo[i]) > 0:
            step_state[i] = WFS_PENDING
        else:
            step_state[i] = WFS_SUCCEEDED


def wf_status_ref(step_state, n_steps, run_active, run_state, counts):
    NR = int(n_steps.shape[0])
    for run in range(NR):
        if not int(run_active[run]):
            continue
        ns = int(n_steps[run])
        vals = [int(step_state[run * 64 + s]) for s in range(ns)]
        if any(v == WFS_FAILED for v in vals):
            run_active[run] = 0
            run_state[run] = WFS_FAILED
            counts[1] += 1
        elif all(v in (WFS_SUCCEEDED, WFS_SKIPPED) for v in vals):
            run_active[run] = 0
            run_state[run] = WFS_SUCCEEDED
            counts[0] += 1


def wf_grant_ref(grant_runs, grant_steps, verdicts, n: int, step_state):
    for i in range(n):
        rs = int(grant_runs[i]) * 64 + int(grant_steps[i])
        if int(step_state[rs]) == WFS_WAITING:
            step_state[rs] = WFS_SUCCEEDED if int(verdicts[i]) else WFS_FAILED


def wf_timeout_scan_ref(step_state, children_out, children_fail,
                        dispatch_tick, tick, cutoff: int, timeout_count):
    tick = int(tick[0]) if hasattr(tick, "__getitem__") else int(tick)
    for i in range(int(step_state.numel())):
        if int(step_state[i]) != WFS_DISPATCHED:
            continue
        lost = int(children_out[i])
        if lost <= 0 or int(dispatch_tick[i]) > tick - cutoff:
            continue
        children_fail[i] += lost
        children_out[i] = 0
        timeout_count[0] += lost


def wf_readmit_ref(run_active, run_state, n_steps, step_state, step_attempts,
                   children_todo, children_out, children_done, children_fail,
                   children_emitted, next_ready, dispatch_tick,
                   todo_tmpl, nready_tmpl, filter_state: int, admit_count):
    NR = int(n_steps.shape[0])
    for run in range(NR):
        if int(run_active[run]) or int(run_state[run]) != filter_state:
            continue
        for s in range(int(n_steps[run])):
            i = run * 64 + s
            step_state[i] = WFS_PENDING
            step_attempts[i] = 0
            children_todo[i] = int(todo_tmpl[i])
            children_out[i] = 0
            children_done[i] = 0
            children_fail[i] = 0
            children_emitted[i] = 0
            next_ready[i] = int(nready_tmpl[i])
            dispatch_tick[i] = 0
        run_state[run] = 0
        run_active[run] = 1
        admit_count[0] += 1
