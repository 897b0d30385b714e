"""HPO helpers (reference: hydragnn/utils/hpo/deephyper.py:15-187 —
DeepHyper/Optuna glue: SLURM node parsing, master-from-host, trial
config generation).  DeepHyper is not in this image; the random-search
fallback keeps config-driven HPO usable offline."""

from __future__ import annotations

import os
import random
from typing import Callable, Dict, List, Optional


def parse_slurm_nodelist(nodelist: str) -> List[str]:
    """Expand SLURM_NODELIST like 'node[001-003,005]' -> hostnames."""
    if "[" not in nodelist:
        return nodelist.split(",")
    prefix, rest = nodelist.split("[", 1)
    rest = rest.rstrip("]")
    hosts = []
    for part in rest.split(","):
        if "-" in part:
            a, b = part.split("-")
            width = len(a)
            for i in range(int(a), int(b) + 1):
                hosts.append(f"{prefix}{i:0{width}d}")
        else:
            hosts.append(f"{prefix}{part}")
    return hosts


def master_from_host(hosts: Optional[List[str]] = None) -> str:
    if hosts:
        return hosts[0]
    nodelist = os.getenv("SLURM_NODELIST")
    if nodelist:
        return parse_slurm_nodelist(nodelist)[0]
    return "127.0.0.1"


def sample_config(space: Dict, rng: random.Random) -> Dict:
    """space: {name: list-of-choices | (lo, hi) | (lo, hi, 'log')}."""
    import math
    out = {}
    for k, v in space.items():
        if isinstance(v, list):
            out[k] = rng.choice(v)
        elif isinstance(v, tuple) and len(v) >= 2:
            lo, hi = v[0], v[1]
            if len(v) == 3 and v[2] == "log":
                out[k] = math.exp(rng.uniform(math.log(lo), math.log(hi)))
            elif isinstance(lo, int) and isinstance(hi, int):
                out[k] = rng.randint(lo, hi)
            else:
                out[k] = rng.uniform(lo, hi)
    return out


def run_random_search(objective: Callable[[Dict], float], space: Dict,
                      num_trials: int = 10, seed: int = 0,
                      maximize: bool = False):
    """Offline HPO driver: random search over `space`, returns
    (best_config, best_value, history)."""
    rng = random.Random(seed)
    best_cfg, best_val = None, None
    history = []
    for _ in range(num_trials):
        cfg = sample_config(space, rng)
        val = objective(cfg)
        history.append((cfg, val))
        better = (best_val is None
                  or (val > best_val if maximize else val < best_val))
        if better:
            best_cfg, best_val = cfg, val
    return best_cfg, best_val, history


def read_node_list() -> List[str]:
    for var in ("SLURM_NODELIST", "LSB_HOSTS", "PBS_NODEFILE"):
        v = os.getenv(var)
        if not v:
            continue
        if var == "PBS_NODEFILE" and os.path.exists(v):
            with open(v) as f:
                return sorted(set(line.strip() for line in f if
                                  line.strip()))
        if var == "LSB_HOSTS":
            return sorted(set(v.split()))
        return parse_slurm_nodelist(v)
    return ["127.0.0.1"]


def read_job_node_list(job_id, job_nodes=None, nodes_per_job: int = 4):
    """Slice the allocation's node list into a per-trial sub-list and
    pick its master host (reference deephyper.py:88)."""
    if job_nodes is None:
        nodelist = os.environ.get("SLURM_JOB_NODELIST", "")
        nodes = parse_slurm_nodelist(nodelist) if nodelist else []
        if not nodes:
            return "127.0.0.1", ""
        start = (int(job_id) * nodes_per_job) % len(nodes)
        job_nodes = nodes[start:start + nodes_per_job]
    return job_nodes[0], ",".join(job_nodes)


def create_ds_config(params, job_id, DEEPHYPER_LOG_DIR="."):
    """Write a per-trial DeepSpeed-style JSON config (reference
    deephyper.py create_ds_config); returns the path."""
    import json
    path = os.path.join(DEEPHYPER_LOG_DIR, f"ds_config_{job_id}.json")
    cfg = {
        "train_batch_size": params.get("batch_size", 32),
        "optimizer": {"type": params.get("optimizer", "AdamW"),
                      "params": {"lr": params.get("lr", 1e-3)}},
    }
    with open(path, "w") as f:
        json.dump(cfg, f, indent=2)
    return path
