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


def get_max_graph_size(dataset) -> int:
    return max(int(d.num_nodes) for d in dataset)


def get_elements_list(dataset):
    """Sorted unique element identities (column 0 of x) across the
    dataset."""
    import torch
    vals = torch.cat([torch.unique(d.x[:, 0]) for d in dataset])
    return torch.unique(vals)


def create_dictionary_from_elements_list(elements_list):
    return {float(e): i for i, e in enumerate(elements_list)}


def create_dataset_categories(dataset):
    """Category id per sample = its composition signature (reference
    compositional_data_splitting.py pattern)."""
    import torch
    elements = get_elements_list(dataset)
    lookup = create_dictionary_from_elements_list(elements)
    categories = []
    for d in dataset:
        counts = [0] * len(lookup)
        vals, n = torch.unique(d.x[:, 0], return_counts=True)
        for v, c in zip(vals.tolist(), n.tolist()):
            counts[lookup[float(v)]] = c
        categories.append(tuple(counts))
    uniq = {c: i for i, c in enumerate(sorted(set(categories)))}
    return [uniq[c] for c in categories]


def duplicate_unique_data_samples(dataset, categories):
    """Ensure every category has >= 2 samples so each split can see
    it (reference behavior: duplicate singletons)."""
    from collections import Counter
    counts = Counter(categories)
    out, out_cat = list(dataset), list(categories)
    for i, c in enumerate(categories):
        if counts[c] == 1:
            out.append(dataset[i])
            out_cat.append(c)
    return out, out_cat


def generate_partition(categories, perc_train: float, seed: int = 0):
    """Per-category proportional train/val/test index partition."""
    import random
    from collections import defaultdict
    rng = random.Random(seed)
    by_cat = defaultdict(list)
    for i, c in enumerate(categories):
        by_cat[c].append(i)
    train, val, test = [], [], []
    for idxs in by_cat.values():
        rng.shuffle(idxs)
        n = len(idxs)
        n_train = max(int(round(n * perc_train)), 1)
        n_val = max((n - n_train) // 2, 0)
        train += idxs[:n_train]
        val += idxs[n_train:n_train + n_val]
        test += idxs[n_train + n_val:]
    return train, val, test


def compositional_stratified_splitting(dataset, perc_train: float):
    """Reference-named entry point: category-stratified split so every
    composition appears in every split where possible."""
    ds, cats = duplicate_unique_data_samples(
        list(dataset), create_dataset_categories(dataset))
    tr, va, te = generate_partition(cats, perc_train)
    pick = lambda idxs: [ds[i] for i in idxs]  # noqa: E731
    return pick(tr), pick(va), pick(te)
