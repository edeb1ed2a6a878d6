#!/bin/bash
# GPU tier (run on an MI355X box, e.g. via gpurun).
set -e
cd "$(dirname "$0")/.."
python -m pytest tests -q -m gpu "$@"
python -c "import __graft_entry__; __graft_entry__.smoke()"
