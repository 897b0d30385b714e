"""Stratified subsampling of graph datasets (reference:
hydragnn/preprocess/stratified_sampling.py:17)."""

from __future__ import annotations

import random
from typing import List


def stratified_sampling(dataset: List, perc: float, seed: int = 0,
                        verbosity: int = 0) -> List:
    """Keep a `perc` fraction, stratified by element composition."""
    from .compositional_splitting import get_keys
    keys = get_keys(dataset)
    by_key = {}
    for i, k in enumerate(keys):
        by_key.setdefault(k, []).append(i)
    rng = random.Random(seed)
    chosen = []
    for k, idxs in by_key.items():
        rng.shuffle(idxs)
        n = max(1, int(len(idxs) * perc))
        chosen.extend(idxs[:n])
    chosen.sort()
    return [dataset[i] for i in chosen]
