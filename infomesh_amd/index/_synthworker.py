"""Torch-free worker for parallel synthetic Zipf draws (spawned
processes import only numpy this way — importing synth.py would pull
torch into every worker, ~2 s each)."""
from __future__ import annotations

import numpy as np


def zipf_chunk(args):
    seed, size, zipf_a, vocab = args
    rng = np.random.default_rng(seed)
    return ((rng.zipf(zipf_a, size=size) - 1) % vocab).astype(np.int64)
