#!/bin/bash
# Flakiness amplifier (reference tests/repeat.sh): run a pytest target N
# times, stop at the first failure. Usage:
#   N=50 scripts/repeat.sh tests/test_many_key_operations.py [pytest args]
set -e
cd "$(dirname "$0")/.."
N=${N:-20}
TARGET=${1:-tests/}
[ $# -gt 0 ] && shift
for i in $(seq 1 "$N"); do
  echo "[repeat] run $i/$N"
  python -m pytest "$TARGET" -q -m "not gpu" "$@"
done
echo "repeat: $N/$N PASSED"
