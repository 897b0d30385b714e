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


def run_deephyper_search(objective: Callable[[Dict], float], space: Dict,
                         num_trials: int = 10, seed: int = 0,
                         maximize: bool = False):
    """DeepHyper CBO search (reference deephyper.py:15-187 uses
    DeepHyper's HpProblem/CBO on the cluster).  Requires deephyper;
    see run_search for the auto-fallback entry point.

    Space grammar matches sample_config: list -> categorical,
    (lo, hi) -> uniform int/float, (lo, hi, 'log') -> log-uniform."""
    from deephyper.hpo import CBO, HpProblem
    try:
        import ConfigSpace.hyperparameters as csh
    except ImportError:
        csh = None

    problem = HpProblem()
    for name, spec in space.items():
        if isinstance(spec, list):
            problem.add_hyperparameter(spec, name)
        elif isinstance(spec, tuple) and len(spec) == 3 \
                and spec[2] == "log":
            if csh is not None:
                problem.add_hyperparameter(
                    csh.UniformFloatHyperparameter(
                        name=name, lower=spec[0], upper=spec[1],
                        log=True), name)
            else:
                problem.add_hyperparameter((spec[0], spec[1]), name)
        else:
            problem.add_hyperparameter((spec[0], spec[1]), name)

    # DeepHyper maximizes; flip the sign for minimization problems.
    def dh_objective(cfg):
        cfg = dict(cfg)
        cfg.pop("job_id", None)
        val = objective(cfg)
        return val if maximize else -val

    search = CBO(problem, dh_objective, random_state=seed)
    results = search.search(max_evals=num_trials)
    # results: DataFrame with 'objective' and 'p:<name>' columns
    objs = results["objective"]
    idx = objs.idxmax()
    best_row = results.loc[idx]
    best_cfg = {c[2:]: best_row[c] for c in results.columns
                if c.startswith("p:")}
    best_val = float(best_row["objective"])
    return best_cfg, (best_val if maximize else -best_val)


def run_search(objective: Callable[[Dict], float], space: Dict,
               num_trials: int = 10, seed: int = 0,
               maximize: bool = False, use_deephyper: bool = False):
    """HPO entry point: DeepHyper CBO when requested and importable,
    offline random search otherwise.  Returns (best_config,
    best_value)."""
    if use_deephyper:
        try:
            import deephyper  # noqa: F401
            return run_deephyper_search(objective, space, num_trials,
                                        seed, maximize)
        except ImportError:
            pass
    best_cfg, best_val, _ = run_random_search(objective, space,
                                              num_trials, seed,
                                              maximize)
    return best_cfg, best_val
