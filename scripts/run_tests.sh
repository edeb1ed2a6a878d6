#!/bin/bash
# CPU test tier (reference tests/run_tests.sh equivalent): the full
# multi-process integration matrix over gloo/loopback.
set -e
cd "$(dirname "$0")/.."
python -m pytest tests -q -m "not gpu" "$@"
