#!/usr/bin/env bash
# Generic topology benchmark runner (reference benchmarks/test_tipc/
# <model>/benchmark_common/run_benchmark.sh:35-171 — drives the real
# trainer for a bounded run, parses `ips:` from the log, SUCCESS/FAIL on
# exit code).
#
# Usage: run_benchmark.sh <config.yaml> <ngpus> "<overrides...>" [max_steps]
set -u
CONFIG=$1
NGPUS=$2
OVERRIDES=${3:-}
MAX_STEPS=${4:-20}
REPO=$(cd "$(dirname "$0")/../../.." && pwd)
LOG=$(mktemp /tmp/tipc_XXXX.log)

CMD_OVR=""
for o in $OVERRIDES; do CMD_OVR="$CMD_OVR -o $o"; done
CMD_OVR="$CMD_OVR -o Engine.max_steps=$MAX_STEPS -o Global.max_steps=$MAX_STEPS"
CMD_OVR="$CMD_OVR -o Engine.logging_freq=5 -o Global.eval_freq= -o Global.save_steps= -o Engine.save_load.save_steps="

if [ "$NGPUS" -gt 1 ]; then
  LAUNCH="python -m torch.distributed.run --nnodes=1 --nproc-per-node $NGPUS --master-addr 127.0.0.1 --master-port ${MASTER_PORT:-29761}"
else
  LAUNCH="python"
fi

echo "[tipc] $LAUNCH tools/train.py -c $CONFIG $CMD_OVR"
(cd "$REPO" && timeout "${TIPC_TIMEOUT:-900}" $LAUNCH tools/train.py -c "$CONFIG" $CMD_OVR) > "$LOG" 2>&1
RC=$?

IPS=$(grep -oE "ips: [0-9.]+ tokens/s" "$LOG" | tail -1)
IMGS=$(grep -oE "ips: [0-9.]+ images/s" "$LOG" | tail -1)
LOSS=$(grep -oE "loss: [0-9.]+" "$LOG" | tail -1)
if [ $RC -eq 0 ]; then
  echo "[tipc] SUCCESS  ${IPS}${IMGS}  ${LOSS}"
else
  echo "[tipc] FAIL rc=$RC — log tail:"
  tail -20 "$LOG"
fi
exit $RC
