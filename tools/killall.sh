#!/usr/bin/env bash
# Stop a training run started by tools/run_pytorch.sh (reference parity:
# tools/killall.sh, which pdsh-killed python on every node by name; here we
# kill exactly the recorded torchrun process group — never by pattern).
set -euo pipefail
cd "$(dirname "$0")/.."
PIDFILE=output/run_pytorch.pid
if [[ ! -f "$PIDFILE" ]]; then
  echo "no $PIDFILE — nothing to stop"
  exit 0
fi
PID="$(cat "$PIDFILE")"
if kill -0 "$PID" 2>/dev/null; then
  # torchrun is a process-group leader; signal the whole group
  kill -- "-$PID" 2>/dev/null || kill "$PID"
  echo "stopped torchrun pid $PID"
else
  echo "pid $PID not running"
fi
rm -f "$PIDFILE"
