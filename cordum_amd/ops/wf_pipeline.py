"""Device-resident workflow engine tick (config #3: 1->256 fan-out +
approval gate over 8 pools, RCCL all-to-all over xGMI).

This is the batched HBM counterpart of the host workflow engine
(cordum_amd/workflow/engine.py; semantics oracle core/workflow/engine.go
:453-827 scheduleReady, :1524-1560 applyResult, :1623-1645
aggregateChildren, :1647-1699 updateRunStatus). Run/step state lives in
per-rank HBM tables; each tick is a fixed kernel sequence:

  wf_sweep    readiness gates (deps, backoff, delay), approval holds,
              condition commits          -> dispatch list of (run, step)
  wf_expand   for_each/worker expansion  -> child-job arena (all-or-nothing
              reservation; arena-full steps retry next sweep)
  K2 routing  children spread over the least-loaded global worker order,
              packed per destination rank (pack_requeue + pack_by_dest with
              the NAK requeue ring), payload gathered from the child arena
  exchange    all_to_all_single over RCCL/xGMI (world > 1)
  echo        device worker pools execute on the receiving rank
  wf_apply    owner-rank result application with deterministic failure
              injection (retry waves), dead-letter application for
              max-deliver drops
  wf_commit   child aggregation, retry + exponential tick backoff, FAILED
              after max_retries
  wf_status   run roll-up -> SUCCEEDED/FAILED counts
  host hop    fresh approval holds drained D2H; the approver (the admin
              analog, gateway.go:3553) grants them next tick H2D

Step kinds: WORKER (one child job), FOR_EACH (fanout children, per-child
retry), APPROVAL (host-granted), CONDITION (pre-evaluated bit ->
SUCCEEDED/SKIPPED), DELAY (tick gate). Dependencies are 64-bit step masks;
SKIPPED satisfies a dependency, FAILED permanently blocks (the run fails).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from . import get_ext
from .pipeline import _RefOps

WFS_PENDING, WFS_DISPATCHED, WFS_WAITING = 0, 1, 2
WFS_SUCCEEDED, WFS_FAILED, WFS_SKIPPED = 3, 4, 5
WFK_WORKER, WFK_FOR_EACH, WFK_APPROVAL, WFK_CONDITION, WFK_DELAY = 0, 1, 2, 3, 4


@dataclass
class StepSpec:
    kind: int
    deps: Sequence[int] = ()
    fanout: int = 1          # children for FOR_EACH (WORKER always 1)
    delay_ticks: int = 0     # DELAY gate
    cond: bool = True        # CONDITION value (pre-evaluated)
    max_parallel: int = 0    # for_each window (0 = unlimited)


@dataclass
class DagSpec:
    """One workflow DAG (<= 64 steps)."""
    steps: List[StepSpec] = field(default_factory=list)

    @classmethod
    def fanout_approval(cls, fanout: int = 256) -> "DagSpec":
        """Config #3 shape: seed -> 1->N fan-out -> approval gate -> final."""
        return cls(steps=[
            StepSpec(WFK_WORKER),
            StepSpec(WFK_FOR_EACH, deps=[0], fanout=fanout),
            StepSpec(WFK_APPROVAL, deps=[1]),
            StepSpec(WFK_WORKER, deps=[2]),
        ])


@dataclass
class WfStats:
    runs_succeeded: int = 0
    runs_failed: int = 0
    ticks: int = 0
    wall_s: float = 0.0


class WorkflowPipeline:
    """Per-rank device workflow engine over a set of runs (same or mixed
    DAGs). World > 1 exchanges children over the padded all-to-all."""

    def __init__(
        self,
        device: torch.device,
        dags: Sequence[DagSpec],
        n_local_workers: int = 256,
        payload_words: int = 64,
        world_size: int = 1,
        rank: int = 0,
        backend: str = "ext",
        fail_ppt: int = 0,
        drop_ppt: int = 0,
        max_retries: int = 2,
        # must exceed the max requeue-ring lifetime (RQ_MAX_DELIVER + 1
        # ticks) so a parked-but-live child is never declared lost — a
        # timeout firing on a child that later applies would double-account
        timeout_cutoff: int = 8,
        approval_verdict: int = 1,
        child_cap: Optional[int] = None,
        pad_cap: Optional[int] = None,
        replicate: Optional[int] = None,
        seed: int = 7,
    ):
        device = torch.device(device)
        self.ext = _RefOps() if backend == "ref" else get_ext(required=True)
        self.device = device
        self.world = world_size
        self.rank = rank
        self.NWL = n_local_workers
        self.W = payload_words
        self.fail_ppt = fail_ppt
        self.drop_ppt = drop_ppt
        self.max_retries = max_retries
        self.timeout_cutoff = timeout_cutoff
        self.approval_verdict = approval_verdict
        if replicate is not None:
            assert len(dags) == 1
        NR = replicate if replicate is not None else len(dags)
        self.NR = NR
        assert all(len(d.steps) <= 64 for d in dags), "device DAGs cap at 64 steps"

        # ---- host-side template tables (the run-creation image) -------------
        TN = 1 if replicate is not None else NR
        st = torch.zeros(TN * 64, dtype=torch.uint8)
        deps = torch.zeros(TN * 64, dtype=torch.int64)
        kinds = torch.zeros(TN * 64, dtype=torch.uint8)
        todo = torch.zeros(TN * 64, dtype=torch.int32)
        maxp = torch.zeros(TN * 64, dtype=torch.int32)
        nready = torch.zeros(TN * 64, dtype=torch.int32)
        nsteps = torch.zeros(TN, dtype=torch.uint8)
        cond = torch.zeros(TN, dtype=torch.int64)
        total_children = 0
        for r, dag in enumerate(dags):
            nsteps[r] = len(dag.steps)
            cbits = 0
            for s, spec in enumerate(dag.steps):
                i = r * 64 + s
                kinds[i] = spec.kind
                m = 0
                for d in spec.deps:
                    m |= 1 << d
                deps[i] = m - (1 << 64) if m >= (1 << 63) else m
                if spec.kind == WFK_FOR_EACH:
                    todo[i] = spec.fanout
                    total_children += spec.fanout
                elif spec.kind == WFK_WORKER:
                    todo[i] = 1
                    total_children += 1
                if spec.kind == WFK_DELAY:
                    nready[i] = spec.delay_ticks
                maxp[i] = spec.max_parallel
                if spec.cond:
                    cbits |= 1 << s
            cond[r] = cbits - (1 << 64) if cbits >= (1 << 63) else cbits
        if replicate is not None:
            st = st.repeat(NR)
            deps = deps.repeat(NR)
            kinds = kinds.repeat(NR)
            todo = todo.repeat(NR)
            maxp = maxp.repeat(NR)
            nready = nready.repeat(NR)
            nsteps = nsteps.repeat(NR)
            cond = cond.repeat(NR)
            total_children *= NR
        self.total_children = total_children
        self._tmpl = {
            "step_state": st, "deps_mask": deps, "step_kind": kinds,
            "children_todo": todo, "next_ready": nready, "n_steps": nsteps,
            "cond_bits": cond, "max_parallel": maxp,
        }

        d = device
        # copy=True is load-bearing: on CPU, .to(device) ALIASES the source,
        # and the live tables must not share storage with the creation
        # template (reset_runs / wf_readmit restore from it)
        self.step_state = st.to(d, copy=True)
        self.deps_mask = deps.to(d, copy=True)
        self.step_kind = kinds.to(d, copy=True)
        self.children_todo = todo.to(d, copy=True)
        self.max_parallel = maxp.to(d, copy=True)
        self.next_ready = nready.to(d, copy=True)
        self.n_steps = nsteps.to(d, copy=True)
        self.cond_bits = cond.to(d, copy=True)
        self.run_active = torch.ones(NR, dtype=torch.uint8, device=d)
        self.run_state = torch.zeros(NR, dtype=torch.uint8, device=d)
        self.step_attempts = torch.zeros(NR * 64, dtype=torch.int32, device=d)
        self.children_out = torch.zeros(NR * 64, dtype=torch.int32, device=d)
        self.children_emitted = torch.zeros(NR * 64, dtype=torch.int32, device=d)
        self.children_done = torch.zeros(NR * 64, dtype=torch.int32, device=d)
        self.children_fail = torch.zeros(NR * 64, dtype=torch.int32, device=d)
        self.wf_counts = torch.zeros(2, dtype=torch.int64, device=d)
        self.dispatch_tick = torch.zeros(NR * 64, dtype=torch.int32, device=d)
        self.timeout_count = torch.zeros(1, dtype=torch.int64, device=d)
        # device tick counter: kernels read it from memory so the whole tick
        # body is hipGraph-capturable (a scalar arg would freeze in capture)
        self.tick_buf = torch.full((1,), -1, dtype=torch.int32, device=d)
        self.retry_count = torch.zeros(1, dtype=torch.int64, device=d)
        self.admit_count = torch.zeros(1, dtype=torch.int64, device=d)

        # dispatch + approval lists (overflowing entries simply retry on the
        # next sweep, so the cap bounds memory, not correctness)
        self.disp_cap = min(NR * 64, 1 << 22)
        self.disp_runs = torch.zeros(self.disp_cap, dtype=torch.int32, device=d)
        self.disp_steps = torch.zeros(self.disp_cap, dtype=torch.int32, device=d)
        self.disp_count = torch.zeros(1, dtype=torch.int32, device=d)
        self.appr_runs = torch.zeros(NR, dtype=torch.int32, device=d)
        self.appr_steps = torch.zeros(NR, dtype=torch.int32, device=d)
        self.appr_count = torch.zeros(1, dtype=torch.int32, device=d)

        # ---- child-job arena -------------------------------------------------
        CB = child_cap or max(1024, min(total_children, 1 << 18))
        self.CB = CB
        self.child_tag = torch.zeros(CB, dtype=torch.int32, device=d)
        self.child_seq = torch.zeros(CB, dtype=torch.int32, device=d)
        self.child_widx = torch.zeros(CB, dtype=torch.int32, device=d)
        self.child_count = torch.zeros(1, dtype=torch.int32, device=d)
        g = torch.Generator().manual_seed(seed * 17 + rank)
        self.child_payload = torch.randint(
            -(1 << 31), (1 << 31) - 1, (CB * payload_words,),
            dtype=torch.int32, generator=g).to(d)

        # ---- worker table (global view over all ranks' pools) ---------------
        NWG = self.NWL * world_size
        self.w_pool = (torch.arange(NWG, dtype=torch.int32) // self.NWL).to(d)
        maxp = max(64, (4 * CB) // max(1, self.NWL))
        self.w_maxp = torch.full((NWG,), maxp, dtype=torch.int32, device=d)
        self.w_labels = torch.zeros(NWG, dtype=torch.int64, device=d)
        self.w_active_local = torch.zeros(self.NWL, dtype=torch.int32, device=d)
        self.w_cpu_local = (torch.rand(self.NWL, generator=g) * 50).to(d)
        self.w_gpu_local = (torch.rand(self.NWL, generator=g) * 50).to(d)
        if world_size == 1:
            # the local view IS the global view: no per-tick D2D copies
            self.w_active = self.w_active_local
            self.w_cpu = self.w_cpu_local
            self.w_gpu = self.w_gpu_local
        else:
            self.w_active = torch.zeros(NWG, dtype=torch.int32, device=d)
            self.w_cpu = torch.zeros(NWG, dtype=torch.float32, device=d)
            self.w_gpu = torch.zeros(NWG, dtype=torch.float32, device=d)
        self.w_keys = torch.zeros(NWG, dtype=torch.int64, device=d)
        self.order_buf = torch.arange(NWG, dtype=torch.int32, device=d)
        self.valid_buf = torch.tensor([NWG], dtype=torch.int32, device=d)

        # ---- padded exchange arenas (child dispatch across ranks) -----------
        self.pad_cap = pad_cap or min(CB, max(64, (4 * CB) // max(1, world_size)))
        cap, Wd, world = self.pad_cap, payload_words, world_size

        def zi(n):
            return torch.zeros(n, dtype=torch.int32, device=d)

        self.pad_send_slots = zi(world * cap)
        self.pad_send_widx = zi(world * cap)
        self.pad_send_cnt = zi(world)
        self.pad_recv_cnt = zi(world)
        self.pad_recv_widx = zi(world * cap)
        self.pad_send_payload = torch.zeros(world * cap * Wd, dtype=torch.int32, device=d)
        self.pad_recv_payload = torch.zeros_like(self.pad_send_payload)
        self.pad_res = torch.zeros_like(self.pad_send_payload)
        self.pad_sums = zi(world * cap)
        self.pad_sums_back = zi(world * cap)
        # requeue ring (children parked on destination overflow) + tag carry.
        # Sized well below the child arena: overflow is the exception, and
        # the ring's payload arena is ping-pong COPIED every tick inside the
        # captured graph — a CB-sized ring made that copy 28% of the tick
        # (profiles/r29_wf_tick_ktrace). Ring-full drops land in the dead
        # list -> wf_apply_dead -> the children retry, so a small ring is
        # safe backpressure, not loss.
        RQ = min(CB, max(4096, CB // 16))
        self.rq_src = zi(RQ)
        self.rq_widx = zi(RQ)
        self.rq_attempts = zi(RQ)
        self.rq_count = zi(1)
        self.rq_dead = torch.zeros(1, dtype=torch.int64, device=d)
        self.rq_payload = torch.zeros(RQ * Wd, dtype=torch.int32, device=d)
        self.rq_tag = zi(RQ)
        self.rq_seq = zi(RQ)
        self.rq_prev_widx = zi(RQ)
        self.rq_prev_attempts = zi(RQ)
        self.rq_prev_count = zi(1)
        self.rq_prev_payload = torch.zeros_like(self.rq_payload)
        self.rq_prev_tag = zi(RQ)
        self.rq_prev_seq = zi(RQ)
        self.dead_src = zi(RQ)
        self.dead_count = zi(1)

        # host-side pinned staging for the approval hop
        pin = d.type == "cuda"
        self._appr_host_runs = torch.zeros(NR, dtype=torch.int32)
        self._appr_host_steps = torch.zeros(NR, dtype=torch.int32)
        if pin:
            self._appr_host_runs = self._appr_host_runs.pin_memory()
            self._appr_host_steps = self._appr_host_steps.pin_memory()
        self._pending_grants: List[tuple] = []
        self._wf_graph = None
        if device.type == "cuda":
            self._appr_h_count = torch.zeros(1, dtype=torch.int32).pin_memory()
            self._appr_ev = torch.cuda.Event()
        else:
            self._appr_h_count = None

        self._tick = 0

    # ---- run admission -------------------------------------------------------
    def reset_runs(self) -> None:
        """Re-admit the whole run wave: copy the creation image back into the
        device tables (this is the honest run-creation H2D cost when the
        template lives on the host; templates are staged once)."""
        if not hasattr(self, "_tmpl_dev"):
            self._tmpl_dev = {k: v.to(self.device) for k, v in self._tmpl.items()}
        t = self._tmpl_dev
        self.step_state.copy_(t["step_state"])
        self.children_todo.copy_(t["children_todo"])
        self.next_ready.copy_(t["next_ready"])
        self.run_active.fill_(1)
        self.run_state.zero_()
        self.step_attempts.zero_()
        self.children_out.zero_()
        self.children_emitted.zero_()
        self.children_done.zero_()
        self.children_fail.zero_()
        self.rq_count.zero_()
        self.rq_prev_count.zero_()
        self.dead_count.zero_()
        self.dispatch_tick.zero_()
        self._pending_grants.clear()
        self._appr_deferred = False
        self._tick = 0  # delay gates + backoff are wave-relative ticks
        self.tick_buf.fill_(-1)

    # ---- collectives ---------------------------------------------------------
    def _heartbeats(self) -> None:
        if self.world > 1:
            dist.all_gather_into_tensor(self.w_active, self.w_active_local)
            if not getattr(self, "_hb_static_done", False):
                dist.all_gather_into_tensor(self.w_cpu, self.w_cpu_local)
                dist.all_gather_into_tensor(self.w_gpu, self.w_gpu_local)
                self._hb_static_done = True
        # world == 1: the global tensors alias the local ones — nothing to do

    def _exchange_out(self) -> None:
        if self.world > 1:
            dist.all_to_all_single(self.pad_recv_cnt, self.pad_send_cnt)
            dist.all_to_all_single(self.pad_recv_widx, self.pad_send_widx)
            dist.all_to_all_single(self.pad_recv_payload, self.pad_send_payload)
        else:
            self.pad_recv_cnt.copy_(self.pad_send_cnt)
            self.pad_recv_widx.copy_(self.pad_send_widx)
            self.pad_recv_payload.copy_(self.pad_send_payload)

    def _exchange_back(self) -> None:
        if self.world > 1:
            dist.all_to_all_single(self.pad_sums_back, self.pad_sums)
        else:
            self.pad_sums_back.copy_(self.pad_sums)

    def _refresh_order(self) -> None:
        if self._tick % 8 != 1:
            return
        self.order_buf.copy_(torch.argsort(self.w_keys).to(torch.int32))
        valid = ((self.w_keys >> 32) & 0xFFFFFFFF).ne(0xFFFFFFFE)
        self.valid_buf.copy_(valid.sum().to(torch.int32).reshape(1))

    # ---- one workflow tick ---------------------------------------------------
    def _tick_device_body(self) -> None:
        """The fixed kernel sequence of one tick (no host decisions): this is
        what gets hipGraph-captured on GPU (world == 1). The device tick
        counter increments in-graph; collectives stay outside (world > 1
        runs the same body eagerly with the exchanges inline)."""
        ext = self.ext
        cap, world, Wd = self.pad_cap, self.world, self.W
        self.tick_buf += 1
        # sweep: readiness -> dispatch list + fresh approval holds
        self.disp_count.zero_()
        self.appr_count.zero_()
        ext.wf_sweep(self.step_state, self.deps_mask, self.n_steps,
                     self.run_active, self.step_kind, self.cond_bits,
                     self.next_ready, self.tick_buf,
                     self.disp_runs, self.disp_steps, self.disp_count,
                     self.appr_runs, self.appr_steps, self.appr_count)

        # expand into the child arena, spread-routed over the global order
        self.child_count.zero_()
        ext.wf_expand(self.disp_runs, self.disp_steps, self.disp_count,
                      self.step_state, self.children_todo, self.children_out,
                      self.max_parallel,
                      self.child_tag, self.child_seq, self.child_widx,
                      self.child_count, self.children_emitted,
                      self.dispatch_tick, self.tick_buf,
                      self.order_buf, self.valid_buf)

        # K2 load view + routing order refresh
        ext.worker_precompute_into(self.w_pool, self.w_active, self.w_maxp,
                                   self.w_cpu, self.w_gpu, self.w_keys)
        self._heartbeats()

        # pack per destination: redeliveries first, then fresh children
        self.rq_prev_widx.copy_(self.rq_widx)
        self.rq_prev_attempts.copy_(self.rq_attempts)
        self.rq_prev_count.copy_(self.rq_count)
        self.rq_prev_payload.copy_(self.rq_payload)
        self.rq_prev_tag.copy_(self.rq_tag)
        self.rq_prev_seq.copy_(self.rq_seq)
        self.rq_count.zero_()
        self.dead_count.zero_()
        self.pad_send_cnt.zero_()
        ext.pack_requeue(self.rq_prev_widx, self.rq_prev_attempts, self.rq_prev_count,
                         self.pad_send_slots, self.pad_send_widx, self.pad_send_cnt,
                         self.NWL, cap,
                         self.rq_src, self.rq_widx, self.rq_attempts,
                         self.rq_count, self.rq_dead,
                         self.dead_src, self.dead_count)
        ext.pack_by_dest(self._child_slots(), self.child_widx, self.child_count,
                         self.pad_send_slots, self.pad_send_widx, self.pad_send_cnt,
                         self.NWL, cap, self.CB,
                         self.rq_src, self.rq_widx, self.rq_attempts,
                         self.rq_count, self.rq_dead,
                         self.dead_src, self.dead_count)
        self.pad_send_cnt.clamp_(max=cap)
        ext.gather_payload_padded(self.child_payload, self.rq_prev_payload,
                                  self.pad_send_slots, self.pad_send_cnt,
                                  self.pad_send_payload, Wd, cap, world)
        # park-survival: payload + tag of newly parked children
        ext.materialize_rq_payload(self.child_payload, self.rq_prev_payload,
                                   self.rq_src, self.rq_count,
                                   self.rq_payload, Wd)
        ext.materialize_rq_payload(self.child_tag, self.rq_prev_tag,
                                   self.rq_src, self.rq_count,
                                   self.rq_tag, 1)
        ext.materialize_rq_payload(self.child_seq, self.rq_prev_seq,
                                   self.rq_src, self.rq_count,
                                   self.rq_seq, 1)

        # dispatch across ranks + device worker execution + results home
        self._exchange_out()
        ext.echo_padded(self.pad_recv_payload, self.pad_recv_cnt, self.pad_res,
                        self.pad_sums, Wd, cap, world)
        self.w_active_local.zero_()
        ext.load_feedback_padded(self.pad_recv_widx, self.pad_recv_cnt,
                                 self.w_active_local, cap, world)
        self._exchange_back()

        # owner-rank application + aggregation + roll-up
        ext.wf_apply(self.pad_send_slots, self.pad_send_cnt, self.child_tag,
                     self.child_seq, self.rq_prev_tag, self.rq_prev_seq,
                     self.children_done, self.children_fail,
                     self.children_out, self.fail_ppt, self.drop_ppt, cap, world)
        ext.wf_apply_dead(self.dead_src, self.dead_count, self.child_tag,
                          self.rq_prev_tag, self.children_fail, self.children_out)
        # K4-WF: steps whose children were lost (worker crash) time out and
        # the remainder retries with backoff (reconciler.go:88-144)
        ext.wf_timeout_scan(self.step_state, self.children_out,
                            self.children_fail, self.dispatch_tick,
                            self.tick_buf, self.timeout_cutoff, self.timeout_count)
        ext.wf_commit(self.step_state, self.step_attempts, self.children_todo,
                      self.children_out, self.children_done, self.children_fail,
                      self.max_parallel,
                      self.next_ready, self.tick_buf, self.max_retries, self.retry_count)
        ext.wf_status(self.step_state, self.n_steps, self.run_active,
                      self.run_state, self.wf_counts)
        if self._appr_h_count is not None:
            # stream-ordered async D2H of the approval count into pinned
            # memory: the host polls it with event.query() instead of a
            # blocking sync every tick (the grant hop already tolerates a
            # tick of lag)
            self._appr_h_count.copy_(self.appr_count, non_blocking=True)

    def _ensure_wf_graph(self) -> bool:
        import os

        if self.device.type != "cuda" or self.world > 1 \
                or os.environ.get("CORDUM_WF_NO_GRAPH"):
            return False
        if self._wf_graph is not None:
            return True
        # eager warmup (allocator steady state), then capture; the warmup
        # ran for real, so restore the run tables (reset) and the cumulative
        # counters (snapshot) afterwards — the triggering tick then replays
        # as tick 0 of a fresh wave
        saved = [(t, t.clone()) for t in
                 (self.wf_counts, self.timeout_count, self.retry_count,
                  self.admit_count, self.rq_dead)]
        for _ in range(2):
            self._tick_device_body()
        torch.cuda.synchronize(self.device)
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._tick_device_body()
            self._wf_graph = g
        except Exception as e:
            import sys

            print(f"[cordum] wf-graph capture failed, running eager: {e!r}",
                  file=sys.stderr)
            self._wf_graph = ()
        self.reset_runs()
        for t, v in saved:
            t.copy_(v)
        torch.cuda.synchronize(self.device)
        return bool(self._wf_graph)

    def _drain_approvals(self) -> None:
        na = int(self._appr_h_count[0]) if self._appr_h_count is not None             else int(self.appr_count.cpu()[0])
        if na > 0:
            na = min(na, self.NR)
            self._pending_grants.append(
                (self.appr_runs[:na].cpu().clone(), self.appr_steps[:na].cpu().clone()))

    def tick(self) -> None:
        ext = self.ext
        self._tick += 1

        # a drain deferred from last tick (its D2H event was still in
        # flight) MUST land before this tick's body overwrites the approval
        # ring — by now the event has long completed, so this rarely blocks
        if getattr(self, "_appr_deferred", False):
            self._appr_ev.synchronize()
            self._drain_approvals()
            self._appr_deferred = False

        # host hop: approvals drained LAST tick are granted now (1-tick
        # admin latency, like the reference's async approve endpoint)
        if self._pending_grants:
            runs, steps = self._pending_grants.pop()
            n = runs.shape[0]
            v = torch.full((n,), self.approval_verdict, dtype=torch.uint8,
                           device=self.device)
            ext.wf_grant(runs.to(self.device), steps.to(self.device), v, n,
                         self.step_state)

        if self.device.type == "cuda" and self.world == 1 and self._ensure_wf_graph():
            self._wf_graph.replay()
        else:
            self._tick_device_body()
        self._refresh_order()

        # drain fresh approval holds for next tick's grant. On GPU the count
        # arrives via the in-body pinned D2H; poll non-blockingly — if the
        # tick is still in flight, defer the drain to the top of the next
        # tick (before the ring is overwritten) instead of stalling here
        if self._appr_h_count is not None:
            self._appr_ev.record()
            if not self._appr_ev.query():
                self._appr_deferred = True
                return
            self._drain_approvals()
        else:
            self._drain_approvals()

    def _child_slots(self):
        """Routable slot list for pack_by_dest = child arena indices 0..C-1."""
        if not hasattr(self, "_iota"):
            self._iota = torch.arange(self.CB, dtype=torch.int32, device=self.device)
        return self._iota

    # ---- continuous mode (config #5) ----------------------------------------
    def _ensure_tmpl_dev(self):
        if not hasattr(self, "_tmpl_dev"):
            self._tmpl_dev = {k: v.to(self.device) for k, v in self._tmpl.items()}
        return self._tmpl_dev

    def readmit_succeeded(self) -> None:
        """Terminal SUCCEEDED runs re-enter from the creation template so the
        table stays full (config #5 'hold 1M concurrent runs')."""
        t = self._ensure_tmpl_dev()
        self.ext.wf_readmit(self.run_active, self.run_state, self.n_steps,
                            self.step_state, self.step_attempts,
                            self.children_todo, self.children_out,
                            self.children_done, self.children_fail,
                            self.children_emitted, self.next_ready,
                            self.dispatch_tick, t["children_todo"],
                            t["next_ready"], WFS_SUCCEEDED, self.admit_count)

    def drain_failed_to_dlq(self, dlq=None) -> int:
        """Host DLQ drain (dlq_store.go analog): record every FAILED run,
        then re-admit those rows through the same template reset. Returns
        the number drained; the caller's DLQ bookkeeping must reconcile
        exactly with the device failed counter at the end of a soak."""
        failed = (self.run_state == WFS_FAILED) & (self.run_active == 0)
        ids = torch.nonzero(failed).flatten()
        n = int(ids.numel())
        if n and dlq is not None:
            from ..store.dlq_store import DLQEntry

            gen = int(self.admit_count.cpu()[0])
            for r in ids.cpu().tolist():
                dlq.add(DLQEntry(job_id=f"run-{r}-g{gen}", topic="job.soak",
                                 status="JOB_STATUS_FAILED",
                                 reason="max retries exceeded (soak)",
                                 reason_code="max_retries_exceeded",
                                 last_state="FAILED",
                                 attempts=self.max_retries))
        if n:
            t = self._ensure_tmpl_dev()
            self.ext.wf_readmit(self.run_active, self.run_state, self.n_steps,
                                self.step_state, self.step_attempts,
                                self.children_todo, self.children_out,
                                self.children_done, self.children_fail,
                                self.children_emitted, self.next_ready,
                                self.dispatch_tick, t["children_todo"],
                                t["next_ready"], WFS_FAILED, self.admit_count)
        return n

    # ---- wave driver ---------------------------------------------------------
    def counts(self) -> tuple:
        c = self.wf_counts.cpu()
        return int(c[0]), int(c[1])

    def active(self) -> int:
        return int(self.run_active.sum().cpu())

    def run_wave(self, max_ticks: int = 256) -> WfStats:
        """Admit the full run wave and tick until every run is terminal."""
        t0 = time.perf_counter()
        self.reset_runs()
        start_ok, start_fail = self.counts()
        ticks = 0
        # world>1: all ranks must tick in lockstep (collectives); keep
        # ticking until every rank's runs are done
        while ticks < max_ticks:
            self.tick()
            ticks += 1
            if ticks % 4 == 0:
                done = self.active() == 0
                if self.world > 1:
                    t = torch.tensor([0 if done else 1],
                                     dtype=torch.int64, device=self.device
                                     if self.device.type == "cuda" else "cpu")
                    dist.all_reduce(t)
                    done = int(t.item()) == 0
                if done:
                    break
        ok, fail = self.counts()
        return WfStats(runs_succeeded=ok - start_ok, runs_failed=fail - start_fail,
                       ticks=ticks, wall_s=time.perf_counter() - t0)
