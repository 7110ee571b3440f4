#!/usr/bin/env bash
# Stop the agentainer-amd server started by start-server.sh (reference
# scripts/stop-server.sh analog). SIGTERM first (graceful: drains, KV
# offload); SIGKILL after 15s — pending WAL requests survive either way
# and replay on the next start.
set -euo pipefail

PIDFILE="${AGENTAINER_PIDFILE:-${TMPDIR:-/tmp}/agentainer-amd.pid}"
if [ ! -f "$PIDFILE" ]; then
    echo "no pidfile at $PIDFILE (server not running?)" >&2
    exit 0
fi
PID="$(cat "$PIDFILE")"
if ! kill -0 "$PID" 2>/dev/null; then
    echo "stale pidfile (pid $PID gone)"
    rm -f "$PIDFILE"
    exit 0
fi
# torchrun puts ranks in their own sessions: signal the children too
CHILDREN="$(ps -o pid= --ppid "$PID" 2>/dev/null || true)"
kill "$PID" 2>/dev/null || true
for c in $CHILDREN; do kill "$c" 2>/dev/null || true; done
for _ in $(seq 1 15); do
    kill -0 "$PID" 2>/dev/null || { rm -f "$PIDFILE"; echo "stopped"; exit 0; }
    sleep 1
done
kill -9 "$PID" 2>/dev/null || true
for c in $CHILDREN; do kill -9 "$c" 2>/dev/null || true; done
rm -f "$PIDFILE"
echo "stopped (forced)"
