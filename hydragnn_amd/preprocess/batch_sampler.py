"""Cost-aware (node-budget) batch samplers.

Reference behavior: hydragnn/preprocess/batch_sampler.py:47-288 —
batches packed greedily under a max-node budget with a deterministic
seed+epoch shuffle; oversized samples handled by policy
(error/single/skip); the distributed variant computes one identical
global plan on every rank, groups batches into equal-cost steps of
world_size, rotates assignment by epoch, and pads so all ranks step the
same number of times.
"""

from __future__ import annotations

import random
from dataclasses import dataclass
from typing import Iterator, List

import torch.distributed as dist


def graph_node_costs(dataset) -> List[int]:
    """Per-sample node counts; uses dataset.get_node_counts() metadata
    when available (DDStore/ADIOS path) instead of materializing
    samples."""
    if hasattr(dataset, "get_node_counts"):
        return list(dataset.get_node_counts())
    return [dataset[i].num_nodes for i in range(len(dataset))]


def _pack(indices: List[int], costs: List[int], max_nodes: int,
          oversized_policy: str, drop_last: bool) -> List[List[int]]:
    batches: List[List[int]] = []
    cur: List[int] = []
    cur_cost = 0
    for i in indices:
        c = costs[i]
        if c > max_nodes:
            if oversized_policy == "error":
                raise ValueError(
                    f"sample {i} has {c} nodes > max_nodes={max_nodes}")
            if oversized_policy == "skip":
                continue
            # "single": its own batch
            if cur:
                batches.append(cur)
                cur, cur_cost = [], 0
            batches.append([i])
            continue
        if cur_cost + c > max_nodes and cur:
            batches.append(cur)
            cur, cur_cost = [], 0
        cur.append(i)
        cur_cost += c
    if cur and not drop_last:
        batches.append(cur)
    return batches


class CostAwareBatchSampler:
    def __init__(self, dataset, max_nodes: int, shuffle: bool = True,
                 seed: int = 0, oversized_policy: str = "error",
                 drop_last: bool = False):
        self.costs = graph_node_costs(dataset)
        self.max_nodes = max_nodes
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0
        self.oversized_policy = oversized_policy
        self.drop_last = drop_last

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def _plan(self) -> List[List[int]]:
        indices = list(range(len(self.costs)))
        if self.shuffle:
            random.Random(self.seed + self.epoch).shuffle(indices)
        return _pack(indices, self.costs, self.max_nodes,
                     self.oversized_policy, self.drop_last)

    def __iter__(self) -> Iterator[List[int]]:
        return iter(self._plan())

    def __len__(self) -> int:
        return len(self._plan())


class DistributedCostAwareBatchSampler(CostAwareBatchSampler):
    """Every rank computes the identical global plan, sorts batches by
    cost, groups them into steps of world_size balanced batches, and
    rotates the rank->batch assignment per epoch; short final steps are
    padded by repeating batches so every rank steps equally."""

    def __init__(self, dataset, max_nodes: int, shuffle: bool = True,
                 seed: int = 0, oversized_policy: str = "error",
                 drop_last: bool = False, num_replicas=None, rank=None):
        super().__init__(dataset, max_nodes, shuffle, seed,
                         oversized_policy, drop_last)
        if num_replicas is None:
            num_replicas = dist.get_world_size() if dist.is_initialized() \
                else 1
        if rank is None:
            rank = dist.get_rank() if dist.is_initialized() else 0
        self.num_replicas = num_replicas
        self.rank = rank

    def _steps(self) -> List[List[List[int]]]:
        batches = self._plan()
        # sort by total cost so each step groups similarly-sized batches
        batches.sort(key=lambda b: sum(self.costs[i] for i in b),
                     reverse=True)
        steps = []
        for s in range(0, len(batches), self.num_replicas):
            group = batches[s:s + self.num_replicas]
            while len(group) < self.num_replicas:  # pad
                group.append(group[len(group) % max(len(group), 1)])
            steps.append(group)
        return steps

    def __iter__(self) -> Iterator[List[int]]:
        steps = self._steps()
        out = []
        for istep, group in enumerate(steps):
            # rotate by epoch + step for load balance determinism
            k = (self.rank + self.epoch + istep) % self.num_replicas
            out.append(group[k])
        return iter(out)

    def __len__(self) -> int:
        return len(self._steps())


@dataclass
class BatchStatistics:
    """Summary of one sampler epoch's batch plan (reference
    batch_sampler.py:25)."""

    num_batches: int
    num_samples: int
    skipped_samples: int
    min_cost: int
    max_cost: int
    mean_cost: float


def compute_batch_statistics(sampler) -> BatchStatistics:
    """Statistics of a CostAwareBatchSampler's current plan."""
    plan = sampler._plan()
    costs = getattr(sampler, "costs", None) or graph_node_costs(
        sampler.dataset)
    batch_costs = [sum(costs[i] for i in b) for b in plan]
    n_used = sum(len(b) for b in plan)
    return BatchStatistics(
        num_batches=len(plan),
        num_samples=n_used,
        skipped_samples=len(costs) - n_used,
        min_cost=min(batch_costs) if batch_costs else 0,
        max_cost=max(batch_costs) if batch_costs else 0,
        mean_cost=(sum(batch_costs) / len(batch_costs)
                   if batch_costs else 0.0))


def graph_node_cost(data) -> int:
    """Cost of a single graph (reference batch_sampler singular
    form); see graph_node_costs for the dataset vectorized variant."""
    return int(data.num_nodes)
