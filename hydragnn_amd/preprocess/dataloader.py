"""HydraDataLoader: thread-prefetching loader with CPU-affinity
pinning (reference: hydragnn/preprocess/load_data.py:54-211 —
ThreadPoolExecutor prefetch, sched_setaffinity, OMP_PLACES parsing;
enabled via HYDRAGNN_CUSTOM_DATALOADER=1)."""

from __future__ import annotations

import os
import queue
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Iterator, List, Optional, Sequence

from ..data import Batch


def parse_omp_places(places: Optional[str] = None) -> List[int]:
    """Parse OMP_PLACES like '{0},{1},{2}' or '{0:4},{8:4}' into a flat
    CPU list (reference load_data.py:54)."""
    places = places if places is not None else os.getenv("OMP_PLACES", "")
    cpus: List[int] = []
    for tok in places.replace(" ", "").split("},"):
        tok = tok.strip("{}")
        if not tok:
            continue
        if ":" in tok:
            start, count = tok.split(":")[:2]
            cpus.extend(range(int(start), int(start) + int(count)))
        elif "-" in tok:
            a, b = tok.split("-")
            cpus.extend(range(int(a), int(b) + 1))
        else:
            try:
                cpus.append(int(tok))
            except ValueError:
                pass
    return cpus


def pin_affinity(width: Optional[int] = None,
                 offset: Optional[int] = None) -> None:
    """Restrict this process to a CPU slice
    (HYDRAGNN_AFFINITY_WIDTH/_OFFSET, reference load_data.py:127)."""
    if not hasattr(os, "sched_setaffinity"):
        return
    width = width if width is not None else \
        int(os.getenv("HYDRAGNN_AFFINITY_WIDTH", "0"))
    offset = offset if offset is not None else \
        int(os.getenv("HYDRAGNN_AFFINITY_OFFSET", "0"))
    if width <= 0:
        return
    local_rank = int(os.getenv("LOCAL_RANK", "0"))
    cpus = parse_omp_places() or list(
        sorted(os.sched_getaffinity(0)))
    start = (offset + local_rank * width) % max(len(cpus), 1)
    chosen = [cpus[(start + i) % len(cpus)] for i in range(width)]
    try:
        os.sched_setaffinity(0, set(chosen))
    except OSError:
        pass


class HydraDataLoader:
    """Batches `dataset` by index lists from `batch_sampler` (or fixed
    batch_size), collating in a thread pool and prefetching
    `prefetch` batches ahead."""

    def __init__(self, dataset, batch_size: int = 32, shuffle: bool = False,
                 batch_sampler=None, num_workers: int = 2,
                 prefetch: int = 4, pin_affinity_width: Optional[int]
                 = None):
        self.dataset = dataset
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.batch_sampler = batch_sampler
        self.num_workers = max(num_workers, 1)
        self.prefetch = prefetch
        if os.getenv("HYDRAGNN_AFFINITY", "0") == "1":
            pin_affinity(pin_affinity_width)
        self._epoch = 0

    def set_epoch(self, epoch: int):
        self._epoch = epoch
        if self.batch_sampler is not None and \
                hasattr(self.batch_sampler, "set_epoch"):
            self.batch_sampler.set_epoch(epoch)

    @property
    def sampler(self):
        return self.batch_sampler

    def _plan(self) -> List[List[int]]:
        if self.batch_sampler is not None:
            return [list(b) for b in self.batch_sampler]
        import random
        idx = list(range(len(self.dataset)))
        if self.shuffle:
            random.Random(self._epoch).shuffle(idx)
        return [idx[i:i + self.batch_size]
                for i in range(0, len(idx), self.batch_size)]

    def _collate(self, indices: Sequence[int]) -> Batch:
        return Batch.from_data_list([self.dataset[i] for i in indices])

    def __len__(self) -> int:
        return len(self._plan())

    def __iter__(self) -> Iterator[Batch]:
        plan = self._plan()
        with ThreadPoolExecutor(max_workers=self.num_workers) as pool:
            futures = queue.Queue()
            it = iter(plan)
            # prime the prefetch window
            for _ in range(min(self.prefetch, len(plan))):
                futures.put(pool.submit(self._collate, next(it)))
            while not futures.empty():
                batch = futures.get().result()
                try:
                    futures.put(pool.submit(self._collate, next(it)))
                except StopIteration:
                    pass
                yield batch
