#!/bin/bash
# Decide the container role from the saved docker-run args:
#   --server-url present -> worker-only (server service stays down)
#   GPUSTACK_AMD_DISABLE_WORKER=1 -> server-only (no embedded worker)
set -e
ARGS_FILE="/run/gpustack-amd/args"
ROLE_DIR="/run/gpustack-amd"
if grep -q -- "--server-url" "$ARGS_FILE" 2>/dev/null; then
    echo worker > "$ROLE_DIR/role"
elif [ "${GPUSTACK_AMD_DISABLE_WORKER:-0}" = "1" ]; then
    echo server > "$ROLE_DIR/role"
else
    echo all > "$ROLE_DIR/role"
fi
echo "[init] role: $(cat $ROLE_DIR/role)"
