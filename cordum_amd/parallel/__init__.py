"""Parallel/distribution layer — the RCCL-over-xGMI fabric.

The reference distributes with NATS queue groups + scheduler replicas
(SURVEY.md §2.3); here distribution is one rank per GPU over
torch.distributed (backend "nccl" == RCCL on ROCm):

 - heartbeat fan-in  -> all_gather_into_tensor of per-rank worker loads
 - job dispatch      -> fixed-capacity padded all_to_all_single
                        (descriptors + payload; per-destination counts are a
                        device-resident vector, so the tick has no host
                        splits sync)
 - result return     -> mirrored all_to_all_single

The collective call sites live in ops/pipeline.py (DevicePipeline's
heartbeat fan-in at :905-908/:1056-1058 and the padded exchanges at
:917-927/:1093-1136, backed by the pack/echo/load-feedback kernels in
ops/hip/cordum_kernels.hip); this package owns process-group bring-up
(init_fabric), used by bench.py and the serve launcher.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist


def init_fabric(backend: str | None = None) -> tuple[int, int, torch.device]:
    """Initialise the per-GPU process group from torchrun env vars.

    Returns (rank, world_size, device). Safe to call with WORLD_SIZE=1 (no
    process group is created)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if torch.cuda.is_available():
        # oversubscription fallback: more ranks than visible GPUs shares
        # devices round-robin (lets a multi-rank job validate on one GPU)
        idx = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(idx)
        device = torch.device(f"cuda:{idx}")
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(
            backend=backend or ("nccl" if device.type == "cuda" else "gloo"),
            rank=rank,
            world_size=world,
        )
    return rank, world, device
