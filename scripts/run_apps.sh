#!/bin/bash
# App smoke runs on toy synthetic data (reference tests/run_apps.sh):
# simple, MF (dsgd + columnwise), word2vec, KGE, CTR — via their tests
# plus a multi-rank simple run through the launcher.
set -e
cd "$(dirname "$0")/.."
python -m pytest tests/test_apps.py tests/test_kge_model.py -q -m "not gpu" "$@"
python -m adapm_amd.launch -n 3 adapm_amd/models/simple.py --iterations 30

# real-file end-to-end: word2vec on a small text corpus (vocab built from
# the file + binary export) and MF on a MatrixMarket ratings file
# (reference run_apps.sh trains on apps/data/ toy files)
TMPD=$(mktemp -d)
trap 'rm -rf "$TMPD"' EXIT
python - "$TMPD" <<'EOF'
import sys, numpy as np
from adapm_amd.models.data_io import write_matrix_market
tmp = sys.argv[1]
rng = np.random.default_rng(0)
vocab = [f"tok{i}" for i in range(80)]
with open(f"{tmp}/corpus.txt", "w") as f:
    for _ in range(300):
        f.write(" ".join(rng.choice(vocab, size=10)) + "\n")
rows = rng.integers(0, 60, size=800); cols = rng.integers(0, 40, size=800)
vals = (rng.standard_normal(800)*0.1 + 1).astype(np.float32)
write_matrix_market(f"{tmp}/ratings.mma", rows, cols, vals, (60, 40))
EOF
python -m adapm_amd.models.word2vec --corpus "$TMPD/corpus.txt" --min-count 1 \
    --dim 16 --epochs 1 --device cpu --binary-output "$TMPD/emb.bin"
python -m adapm_amd.models.mf --data "$TMPD/ratings.mma" --rank 8 --epochs 2 \
    --schedule plain_sgd --device cpu
echo "run_apps: PASSED"
