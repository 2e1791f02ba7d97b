#!/usr/bin/env bash
# 8×MI355X single-node launch: one gateway rank per GPU over RCCL/xGMI
# (BASELINE config 3). Works first-try on a fresh node:
#   ./deploy/launch_8gpu.sh bench            # flagship HTTP bench, N=8
#   ./deploy/launch_8gpu.sh bench-engine     # engine-only bench, N=8
#   ./deploy/launch_8gpu.sh serve            # 8 serving ranks, ports 4444..4451
#   ./deploy/launch_8gpu.sh test             # world-8 fabric smoke
set -euo pipefail
cd "$(dirname "$0")/.."

export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}  # dmabuf IPC (pool hosts)
export MASTER_ADDR=127.0.0.1
NPROC=${NPROC:-8}

run() {
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NPROC" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" "$@"
}

case "${1:-bench}" in
  bench)        shift || true; run bench.py --gpus "$NPROC" "$@" ;;
  bench-engine) shift || true; run bench.py --gpus "$NPROC" --mode engine "$@" ;;
  serve)        shift || true; run -m mcp_context_forge_amd serve --distributed "$@" ;;
  test)         shift || true; NPROC=$NPROC run -m pytest tests/test_rccl_silicon.py -x -q -s "$@" ;;
  *) echo "usage: $0 {bench|bench-engine|serve|test} [extra args]" >&2; exit 2 ;;
esac
