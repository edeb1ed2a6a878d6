#!/bin/bash
# App smoke runs on toy synthetic data (reference tests/run_apps.sh):
# simple, MF (dsgd + columnwise), word2vec, KGE, CTR — via their tests
# plus a multi-rank simple run through the launcher.
set -e
cd "$(dirname "$0")/.."
python -m pytest tests/test_apps.py tests/test_kge_model.py -q -m "not gpu" "$@"
python -m adapm_amd.launch -n 3 adapm_amd/models/simple.py --iterations 30
echo "run_apps: PASSED"
