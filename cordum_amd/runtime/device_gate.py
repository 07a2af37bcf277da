"""Batched device safety gate: SafetyKernel.evaluate vectorized through K1.

The serving-path seam the north star requires: API-submitted jobs are encoded
into bitset descriptors (ops/policy_compile.JobEncoder), evaluated by the K1
first-match kernel on the GPU, and the matched rule index is expanded back
into the exact PolicyCheckResponse the host kernel would have produced —
the pre-gates (topic must start `job.`, kernel.go:129-140) and the post-match
logic (tenant MCP gate, effective-config restrictions, response assembly)
run through the SAME code as SafetyKernel.evaluate (finish_response), so host
and device paths cannot drift.

Policy hot-reload: subscribes to SafetyKernel.on_policy_swap and recompiles
the rule tensors (words 1->2->4 until the vocabulary fits). A policy whose
vocabulary overflows compiles exact=False and the gate falls back to the host
evaluator per job — correctness never depends on the compile succeeding.
"""
from __future__ import annotations

import threading
from typing import List, Optional

import torch

from ..ops.policy_compile import CompiledPolicy, JobEncoder, compile_policy
from ..protocol.capv2 import DecisionType, PolicyCheckRequest, PolicyCheckResponse
from ..safety import policy as pol
from ..safety.kernel import SafetyKernel


class DeviceBatchGate:
    def __init__(self, kernel: SafetyKernel, device: torch.device, ext):
        self.kernel = kernel
        self.device = torch.device(device)
        self.ext = ext
        self._mu = threading.Lock()
        self._compiled: Optional[CompiledPolicy] = None
        self._cpol: Optional[CompiledPolicy] = None  # device copy
        self._encoder: Optional[JobEncoder] = None
        self._policy: Optional[pol.SafetyPolicy] = None
        self._snapshot = ""
        self.batches_evaluated = 0
        self.jobs_evaluated = 0
        kernel.on_policy_swap(self._recompile)

    # -- policy swap -------------------------------------------------------------
    def _recompile(self, policy: Optional[pol.SafetyPolicy], snapshot: str) -> None:
        compiled = None
        if policy is not None:
            for w in (1, 2, 4):
                c = compile_policy(policy, words=w, snapshot=snapshot)
                if c.exact:
                    compiled = c
                    break
            else:
                compiled = None  # vocabulary overflow: host fallback
        with self._mu:
            self._policy = policy
            self._snapshot = snapshot
            self._compiled = compiled
            self._cpol = compiled.to(self.device) if compiled is not None else None
            self._encoder = JobEncoder(compiled) if compiled is not None else None

    @property
    def device_active(self) -> bool:
        with self._mu:
            return self._cpol is not None

    # -- batched evaluate ----------------------------------------------------------
    def evaluate_batch(self, reqs: List[PolicyCheckRequest]) -> List[PolicyCheckResponse]:
        """One K1 launch for the whole submit batch; per-job pre/post logic is
        the host kernel's own code. Falls back to per-job host evaluate when
        the policy didn't compile exactly."""
        with self._mu:
            cpol, encoder = self._cpol, self._encoder
            policy, snapshot = self._policy, self._snapshot

        out: List[Optional[PolicyCheckResponse]] = [None] * len(reqs)
        idxs: List[int] = []
        inputs: List[pol.PolicyInput] = []
        topics: List[str] = []
        for i, req in enumerate(reqs):
            topic = (req.topic or "").strip()
            if not topic:
                out[i] = PolicyCheckResponse(decision=DecisionType.DENY, reason="missing topic")
                continue
            if not topic.startswith("job."):
                out[i] = PolicyCheckResponse(decision=DecisionType.DENY, reason="unsupported topic")
                continue
            if cpol is None:
                out[i] = self.kernel.evaluate(req)
                continue
            idxs.append(i)
            inputs.append(self.kernel.input_from_request(req))
            topics.append(topic)

        if idxs:
            jb = encoder.encode(inputs).to(self.device)
            first = self.ext.policy_first_match(
                cpol.any_masks, cpol.all_masks, cpol.secrets,
                cpol.mcp_masks, cpol.mcp_any,
                jb.any_bits, jb.all_bits, jb.secrets, jb.mcp_bits, jb.mcp_used, 0,
            )
            first_host = first.cpu().tolist()  # one D2H for the batch
            R = cpol.n_rules
            for j, i in enumerate(idxs):
                ri = first_host[j]
                if 0 <= ri < R:
                    pd = pol.decision_from_rule(cpol.rules[ri])
                else:  # no match (or unprimed INT_MAX) = default allow
                    pd = pol.PolicyDecision(decision=pol.DECISION_ALLOW)
                out[i] = self.kernel.finish_response(
                    pd, inputs[j], topics[j], reqs[i], policy, snapshot
                )
            self.batches_evaluated += 1
            self.jobs_evaluated += len(idxs)
        return out  # type: ignore[return-value]
