"""Compositional stratified splitting (reference: hydragnn/utils/
datasets/compositional_data_splitting.py:19-156): categorize samples by
element composition and stratify the train/val/test split over the
categories with sklearn StratifiedShuffleSplit; falls back to a plain
shuffle when a category is too small to stratify."""

from __future__ import annotations

import random
from typing import List


def get_keys(dataset) -> List[int]:
    keys = []
    for i in range(len(dataset)):
        d = dataset[i]
        z = d.get("z")
        if z is None:
            x = d.get("x")
            z = x[:, 0].long() if x is not None else None
        if z is None:
            keys.append(0)
        else:
            elems = tuple(sorted(set(int(v) for v in z.flatten())))
            keys.append(hash(elems) % (2 ** 31))
    return keys


def compositional_stratified_split(dataset, perc_train: float,
                                   seed: int = 0):
    n = len(dataset)
    keys = get_keys(dataset)
    perc_val = (1.0 - perc_train) / 2.0
    try:
        from sklearn.model_selection import StratifiedShuffleSplit
        import numpy as np
        ycat = np.asarray(keys)
        # collapse singleton categories (stratify needs >=2 per class)
        vals, counts = np.unique(ycat, return_counts=True)
        singles = set(vals[counts < 3].tolist())
        ycat = np.array([-1 if k in singles else k for k in ycat])
        idx = np.arange(n)
        sss = StratifiedShuffleSplit(n_splits=1, train_size=perc_train,
                                     random_state=seed)
        train_idx, rest_idx = next(sss.split(idx, ycat))
        rest_frac = perc_val / (1.0 - perc_train)
        sss2 = StratifiedShuffleSplit(n_splits=1, train_size=rest_frac,
                                      random_state=seed)
        try:
            val_rel, test_rel = next(sss2.split(rest_idx, ycat[rest_idx]))
            val_idx = rest_idx[val_rel]
            test_idx = rest_idx[test_rel]
        except ValueError:
            half = len(rest_idx) // 2
            val_idx, test_idx = rest_idx[:half], rest_idx[half:]
    except (ImportError, ValueError):
        idx = list(range(n))
        random.Random(seed).shuffle(idx)
        n_train = int(n * perc_train)
        n_val = int(n * perc_val)
        train_idx = idx[:n_train]
        val_idx = idx[n_train:n_train + n_val]
        test_idx = idx[n_train + n_val:]
    take = lambda ids: [dataset[int(i)] for i in ids]
    return take(train_idx), take(val_idx), take(test_idx)
