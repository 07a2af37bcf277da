"""cordum_amd — an MI355X-native control plane for autonomous agent workflows.

A from-scratch rebuild of the capabilities of cordum-io/cordum (Go + NATS + Redis)
as a single-process framework for one 8-GPU AMD MI355X node:

- run/job/pointer state lives in HBM (torch tensors / device arenas) instead of Redis
- job dispatch between per-GPU worker pools travels over RCCL (torch.distributed
  "nccl" backend == RCCL on ROCm) across xGMI instead of NATS JetStream
- the hot paths (batched policy rule x job evaluation, least-loaded worker scoring,
  job state transitions, deadline scans) are hand-written CDNA4 HIP kernels
- the HTTP API surface, CAP v2 message shapes, cordumctl CLI, pack format and the
  YAML config formats stay compatible with the reference.

Layer map mirrors SURVEY.md §1; reference citations in docstrings are
`path:line` into the Go reference tree.
"""

__version__ = "0.1.0"
