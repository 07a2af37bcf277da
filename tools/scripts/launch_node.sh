#!/usr/bin/env bash
# Single-node 8-GPU launch (replaces the reference's
# docker-compose five-service stack with one node).
#
# Two planes:
#  1) the serving/control plane: one `cordumctl serve` process (gateway +
#     scheduler + workflow engine + safety kernel in-process), device
#     dispatch on GPU 0 when available, durable WAL under --state-dir;
#  2) the batched data plane across all GPUs (one rank per GPU over
#     RCCL/xGMI), exercised via bench.py --gpus N (the driver's scaling
#     entrypoint) or your own WorkflowPipeline/DevicePipeline embedding.
#
# Usage:
#   tools/scripts/launch_node.sh serve [PORT] [STATE_DIR]
#   tools/scripts/launch_node.sh bench [NGPUS] [EXTRA ARGS...]
set -euo pipefail
cd "$(dirname "$0")/../.."
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}

case "${1:-serve}" in
  serve)
    PORT="${2:-8081}"
    STATE="${3:-$HOME/.cordum/state}"
    exec python -m cordum_amd.cli.cordumctl up --port "$PORT" --state-dir "$STATE"
    ;;
  bench)
    N="${2:-8}"
    shift 2 || shift $#
    exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port 29650 \
      bench.py --gpus "$N" "$@"
    ;;
  *)
    echo "usage: $0 serve [port] [state_dir] | bench [ngpus] [args...]" >&2
    exit 2
    ;;
esac
