"""Real-data ingestion for the example apps (parity with the reference's
file readers; synthetic generators remain the bench path since the
environment has no network for datasets):

- word2vec corpus vocabulary build + tokenized sentence stream
  (reference apps/word2vec.cc:147-320 ReadWord/LearnVocabFromTrainFile)
- word2vec binary checkpoint format (word2vec.cc:367-416 writes both
  binary and text; text export lives on the model)
- MatrixMarket coordinate reader for MF ratings (reference apps/mf/io.h
  reads .mma files)
"""
from __future__ import annotations

import numpy as np

SENT_BOUNDARY = "</s>"


def build_vocab(corpus_path: str, min_count: int = 5, max_vocab: int = None):
    """Whitespace-tokenized vocabulary, newline = sentence boundary.
    Mirrors reference LearnVocabFromTrainFile (word2vec.cc:147-320):
    index 0 is the sentence-boundary token `</s>`, the rest sorted by
    descending count; words under min_count are discarded.

    Returns (words: list[str], counts: np.ndarray[int64], word2id: dict).
    """
    from collections import Counter

    counts = Counter()
    n_lines = 0
    with open(corpus_path, "r", errors="replace") as f:
        for line in f:
            n_lines += 1
            counts.update(line.split())
    items = [(w, c) for w, c in counts.items() if c >= min_count and w != SENT_BOUNDARY]
    items.sort(key=lambda wc: (-wc[1], wc[0]))
    if max_vocab is not None and len(items) > max_vocab - 1:
        items = items[: max_vocab - 1]
    words = [SENT_BOUNDARY] + [w for w, _ in items]
    cnt = np.array([max(1, n_lines)] + [c for _, c in items], dtype=np.int64)
    word2id = {w: i for i, w in enumerate(words)}
    return words, cnt, word2id


def read_sentences(corpus_path: str, word2id: dict, max_sentence_len: int = 1000):
    """Yield sentences as int64 word-id arrays; unknown words are skipped
    (reference ReadWordIndex returns -1 and the trainer skips), long
    lines are split at max_sentence_len (reference MAX_SENTENCE_LENGTH)."""
    with open(corpus_path, "r", errors="replace") as f:
        for line in f:
            ids = [word2id[w] for w in line.split() if w in word2id]
            for i in range(0, len(ids), max_sentence_len):
                chunk = ids[i:i + max_sentence_len]
                if chunk:
                    yield np.asarray(chunk, dtype=np.int64)


def export_word2vec_binary(path: str, words, vectors: np.ndarray):
    """Classic word2vec binary format (word2vec.cc:367-380): header
    '<vocab> <dim>\\n', then per word 'word ' + dim float32 LE + '\\n'."""
    vecs = np.ascontiguousarray(vectors, dtype=np.float32)
    assert len(words) == vecs.shape[0]
    with open(path, "wb") as f:
        f.write(f"{len(words)} {vecs.shape[1]}\n".encode())
        for w, v in zip(words, vecs):
            f.write(w.encode() + b" ")
            f.write(v.tobytes())
            f.write(b"\n")


def read_word2vec_binary(path: str):
    """Inverse of export_word2vec_binary; returns (words, vectors)."""
    with open(path, "rb") as f:
        header = f.readline().split()
        n, dim = int(header[0]), int(header[1])
        words, vecs = [], np.empty((n, dim), dtype=np.float32)
        for i in range(n):
            w = bytearray()
            while True:
                c = f.read(1)
                if c == b" ":
                    break
                if not c:
                    raise ValueError("truncated word2vec binary file")
                w.extend(c)
            words.append(w.decode())
            vecs[i] = np.frombuffer(f.read(4 * dim), dtype="<f4")
            f.read(1)  # trailing newline
    return words, vecs


def read_matrix_market(path: str):
    """MatrixMarket coordinate reader (the reference MF app reads .mma
    ratings, apps/mf/io.h). Supports 'coordinate real/integer/pattern
    general'. Returns (rows, cols, vals, (M, N)) with 0-based indices."""
    with open(path, "r") as f:
        header = f.readline()
        if not header.startswith("%%MatrixMarket"):
            raise ValueError(f"{path}: not a MatrixMarket file")
        parts = header.split()
        if len(parts) < 4 or parts[1] != "matrix" or parts[2] != "coordinate":
            raise ValueError(f"{path}: only 'matrix coordinate' is supported")
        field = parts[3]
        if field not in ("real", "integer", "pattern"):
            raise ValueError(f"{path}: unsupported field '{field}'")
        line = f.readline()
        while line.startswith("%"):
            line = f.readline()
        m, n, nnz = (int(x) for x in line.split())
        rows = np.empty(nnz, dtype=np.int64)
        cols = np.empty(nnz, dtype=np.int64)
        vals = np.ones(nnz, dtype=np.float32)
        for i in range(nnz):
            toks = f.readline().split()
            rows[i] = int(toks[0]) - 1
            cols[i] = int(toks[1]) - 1
            if field != "pattern":
                vals[i] = float(toks[2])
    return rows, cols, vals, (m, n)


def write_matrix_market(path: str, rows, cols, vals, shape):
    """Writer (reference apps can emit factors/ratings in .mma form)."""
    rows = np.asarray(rows)
    with open(path, "w") as f:
        f.write("%%MatrixMarket matrix coordinate real general\n")
        f.write(f"{shape[0]} {shape[1]} {len(rows)}\n")
        for i, j, v in zip(rows, cols, vals):
            f.write(f"{int(i) + 1} {int(j) + 1} {float(v):.7g}\n")
