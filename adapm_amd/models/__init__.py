from . import ctr, kge, mf, simple, word2vec  # noqa: F401
