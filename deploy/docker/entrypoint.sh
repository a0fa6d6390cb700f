#!/bin/bash
# Save `docker run` args for the s6 services (reference entrypoint.sh
# behavior: args decide the role; no args = all-in-one server+worker),
# then hand off to the s6-overlay supervision tree.
set -e
ARGS_FILE="/run/gpustack-amd/args"
mkdir -p "$(dirname "$ARGS_FILE")"
: > "$ARGS_FILE"
for arg in "$@"; do
    printf '%s\n' "$arg" >> "$ARGS_FILE"
done
# raise nofile for many-connection serving, best effort
ulimit -n 65535 2>/dev/null || true
exec /init
