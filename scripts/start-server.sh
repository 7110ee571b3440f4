#!/usr/bin/env bash
# Start the agentainer-amd server (reference scripts/start-server.sh:36-48
# analog — no Docker/Redis side-cars: the engine and store are in-process).
#
#   scripts/start-server.sh                 # single GPU (or CPU fallback)
#   TP_DEGREE=8 scripts/start-server.sh     # tensor-parallel over 8 GPUs
#
# Environment (all optional):
#   AGENTAINER_SERVER_PORT   REST port            (default 8081)
#   AGENTAINER_STORE_PATH    state root           (default ~/.agentainer_amd)
#   AGENTAINER_ENGINE_DEVICE cuda|cpu|echo|auto   (default auto)
#   TP_DEGREE                ranks (torchrun)     (default 1)
set -euo pipefail

ROOT="$(cd "$(dirname "${BASH_SOURCE[0]}")/.." && pwd)"
PIDFILE="${AGENTAINER_PIDFILE:-${TMPDIR:-/tmp}/agentainer-amd.pid}"
TP="${TP_DEGREE:-1}"
PORT="${AGENTAINER_SERVER_PORT:-8081}"

if [ -f "$PIDFILE" ] && kill -0 "$(cat "$PIDFILE")" 2>/dev/null; then
    echo "server already running (pid $(cat "$PIDFILE"))" >&2
    exit 1
fi

# the RCCL/xGMI multi-process path requires dmabuf IPC on this driver
export HSA_ENABLE_IPC_MODE_LEGACY="${HSA_ENABLE_IPC_MODE_LEGACY:-0}"
cd "$ROOT"

if [ "$TP" -gt 1 ]; then
    nohup python -m torch.distributed.run --nnodes=1 --nproc-per-node "$TP" \
        --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29531}" \
        -m agentainer_amd.tp_serve \
        > "${AGENTAINER_LOG:-${TMPDIR:-/tmp}/agentainer-amd.log}" 2>&1 &
else
    nohup python -m agentainer_amd.cli server \
        > "${AGENTAINER_LOG:-${TMPDIR:-/tmp}/agentainer-amd.log}" 2>&1 &
fi
echo $! > "$PIDFILE"

for _ in $(seq 1 120); do
    if curl -fsS "http://127.0.0.1:${PORT}/health" >/dev/null 2>&1; then
        echo "agentainer-amd server up on :${PORT} (pid $(cat "$PIDFILE"))"
        exit 0
    fi
    if ! kill -0 "$(cat "$PIDFILE")" 2>/dev/null; then
        echo "server process died; see ${AGENTAINER_LOG:-${TMPDIR:-/tmp}/agentainer-amd.log}" >&2
        rm -f "$PIDFILE"
        exit 1
    fi
    sleep 1
done
echo "server did not become healthy within 120s" >&2
exit 1
