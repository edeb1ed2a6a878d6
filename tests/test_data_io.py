"""Real-data ingestion parity (reference word2vec.cc:147-320 vocab build,
:367-416 binary export; apps/mf/io.h MatrixMarket): vocab build from a
corpus file, word2vec binary round-trip, .mma round-trip, and end-to-end
training of both apps on tiny real files."""
import os
import subprocess
import sys

import numpy as np

from adapm_amd.models.data_io import (build_vocab, export_word2vec_binary,
                                      read_matrix_market, read_sentences,
                                      read_word2vec_binary, write_matrix_market)

CORPUS = """the quick brown fox jumps over the lazy dog
the dog barks at the fox
a quick fox and a lazy dog
the the the quick quick dog
"""


def test_vocab_build(tmp_path):
    p = tmp_path / "corpus.txt"
    p.write_text(CORPUS)
    words, counts, w2id = build_vocab(str(p), min_count=2)
    assert words[0] == "</s>"
    assert counts[0] == 4  # newlines
    # sorted by descending count after </s>
    assert words[1] == "the" and counts[1] == 7
    body = {w: c for w, c in zip(words[1:], counts[1:])}
    assert body["quick"] == 4 and body["dog"] == 4 and body["fox"] == 3
    assert "barks" not in body  # min_count=2 filters singletons
    sents = list(read_sentences(str(p), w2id))
    assert len(sents) == 4
    # unknown (filtered) words are skipped
    assert all(int(i) < len(words) for s in sents for i in s)
    assert [words[i] for i in sents[1]] == ["the", "dog", "the", "fox"]


def test_word2vec_binary_roundtrip(tmp_path):
    p = tmp_path / "emb.bin"
    words = ["</s>", "hello", "world"]
    vecs = np.random.default_rng(0).standard_normal((3, 8)).astype(np.float32)
    export_word2vec_binary(str(p), words, vecs)
    w2, v2 = read_word2vec_binary(str(p))
    assert w2 == words
    np.testing.assert_array_equal(v2, vecs)


def test_matrix_market_roundtrip(tmp_path):
    p = tmp_path / "r.mma"
    rng = np.random.default_rng(1)
    rows = rng.integers(0, 50, size=200)
    cols = rng.integers(0, 30, size=200)
    vals = rng.standard_normal(200).astype(np.float32)
    write_matrix_market(str(p), rows, cols, vals, (50, 30))
    r2, c2, v2, shape = read_matrix_market(str(p))
    assert shape == (50, 30)
    np.testing.assert_array_equal(r2, rows)
    np.testing.assert_array_equal(c2, cols)
    np.testing.assert_allclose(v2, vals, rtol=1e-5)


def test_w2v_app_on_real_corpus(tmp_path):
    corpus = tmp_path / "corpus.txt"
    rng = np.random.default_rng(3)
    vocab = [f"word{i}" for i in range(50)]
    lines = [" ".join(rng.choice(vocab, size=12)) for _ in range(200)]
    corpus.write_text("\n".join(lines) + "\n")
    out_bin = tmp_path / "emb.bin"
    env = dict(os.environ, RANK="0", WORLD_SIZE="1")
    r = subprocess.run(
        [sys.executable, "-m", "adapm_amd.models.word2vec", "--corpus", str(corpus),
         "--min-count", "1", "--dim", "16", "--epochs", "1", "--device", "cpu",
         "--binary-output", str(out_bin)],
        capture_output=True, text=True, timeout=300, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-2000:]
    words, vecs = read_word2vec_binary(str(out_bin))
    assert len(words) == 51 and vecs.shape[1] == 16  # 50 words + </s>
    assert np.isfinite(vecs).all()


def test_mf_app_on_mma_file(tmp_path):
    mma = tmp_path / "ratings.mma"
    rng = np.random.default_rng(5)
    rows = rng.integers(0, 40, size=500)
    cols = rng.integers(0, 25, size=500)
    vals = (rng.standard_normal(500) * 0.1 + 1.0).astype(np.float32)
    write_matrix_market(str(mma), rows, cols, vals, (40, 25))
    env = dict(os.environ, RANK="0", WORLD_SIZE="1")
    r = subprocess.run(
        [sys.executable, "-m", "adapm_amd.models.mf", "--data", str(mma),
         "--rank", "8", "--epochs", "2", "--schedule", "plain_sgd", "--device", "cpu"],
        capture_output=True, text=True, timeout=300, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr[-2000:]
    assert "epoch 1" in r.stdout
