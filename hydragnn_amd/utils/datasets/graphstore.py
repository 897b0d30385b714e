"""Binary graph store: the MI355X-native replacement for the
ADIOS2/DDStore data plane.

API parity targets:
  AdiosWriter.save / AdiosDataset(read modes preload | mmap) ->
  GraphStoreWriter / GraphStoreDataset
  DistDataset (DDStore RDMA get) -> DistDataset shim with
  epoch_begin/epoch_end + get-by-offset over the same store.

Design: per-key flat binary .npy files (memmap-read) + per-sample
count/offset metadata + a JSON meta file carrying attributes (minmax,
pna_deg, dataset names).  Single-node MI355X has 288 GB HBM x 8 + host
RAM: the page cache IS the shared store, so sample fetches are memmap
reads — no RDMA layer needed; multi-node would swap the fetch for a
one-sided read without changing this interface.
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch
import torch.distributed as dist

from ...data import Data
from .abstractbasedataset import AbstractBaseDataset


def _rank_world():
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


class GraphStoreWriter:
    """Collects Data samples (rank-local) and writes per-key global
    arrays with variable-dim count metadata."""

    def __init__(self, label: str, basedir: str, comm=None):
        self.label = label
        self.basedir = basedir
        self.samples: List[Data] = []
        self.attributes: Dict[str, object] = {}

    def add(self, label_or_samples, samples=None):
        items = samples if samples is not None else label_or_samples
        if isinstance(items, Data):
            items = [items]
        self.samples.extend(items)

    def add_global(self, name: str, value) -> None:
        if torch.is_tensor(value):
            value = value.tolist()
        elif isinstance(value, np.ndarray):
            value = value.tolist()
        self.attributes[name] = value

    def save(self) -> None:
        rank, world = _rank_world()
        os.makedirs(self.basedir, exist_ok=True)
        keys = set()
        for s in self.samples:
            for k, v in s.items():
                if torch.is_tensor(v):
                    keys.add(k)
        keys.discard("num_nodes_")
        meta = {"keys": {}, "attrs": self.attributes,
                "nranks": world, "label": self.label,
                "ndata_rank": len(self.samples)}
        for k in sorted(keys):
            arrays = []
            counts = []
            shapes_tail = None
            for s in self.samples:
                v = s.get(k)
                if v is None:
                    counts.append(0)
                    continue
                a = v.detach().cpu().numpy()
                if a.ndim == 0:
                    a = a.reshape(1)
                if k == "edge_index":
                    # store [E, 2] so the variable dim leads
                    a = np.ascontiguousarray(a.T)
                if a.size == 0:
                    counts.append(0)
                    continue
                arrays.append(a.reshape(a.shape[0], -1))
                counts.append(a.shape[0])
                shapes_tail = list(a.shape[1:])
                meta.setdefault("ndims", {})[k] = a.ndim
            flat = (np.concatenate(arrays, axis=0) if arrays
                    else np.zeros((0, 1)))
            np.save(os.path.join(
                self.basedir, f"{self.label}-{k}.r{rank}.npy"), flat)
            meta["keys"][k] = {
                "dtype": str(flat.dtype),
                "tail_shape": shapes_tail if shapes_tail is not None
                              else [1],
                "counts": counts,
            }
        with open(os.path.join(self.basedir,
                               f"{self.label}-meta.r{rank}.json"),
                  "w") as f:
            json.dump(meta, f)
        if dist.is_initialized():
            dist.barrier()


class GraphStoreDataset(AbstractBaseDataset):
    """Reads a GraphStore: preload=True materializes samples in memory,
    otherwise per-sample memmap reads (the node-local shared-memory
    read mode of the reference)."""

    def __init__(self, basedir: str, label: str = "total",
                 preload: bool = False, subset: Optional[Sequence[int]]
                 = None, comm=None):
        super().__init__()
        self.basedir = basedir
        self.label = label
        self.preload = preload
        # discover rank shards
        metas = []
        r = 0
        while True:
            p = os.path.join(basedir, f"{label}-meta.r{r}.json")
            if not os.path.exists(p):
                break
            with open(p) as f:
                metas.append(json.load(f))
            r += 1
        if not metas:
            raise FileNotFoundError(
                f"no GraphStore meta under {basedir} for label {label}")
        self.metas = metas
        for k, v in metas[0].get("attrs", {}).items():
            setattr(self, k, v)
        # build global sample table: (shard, local_idx)
        self.table = []
        for ishard, m in enumerate(metas):
            for i in range(m["ndata_rank"]):
                self.table.append((ishard, i))
        # per-key offsets per shard
        self.key_info = {}
        self.mmaps = {}
        for k, info in metas[0]["keys"].items():
            per_shard = []
            for ishard, m in enumerate(metas):
                counts = m["keys"][k]["counts"]
                offsets = np.zeros(len(counts) + 1, dtype=np.int64)
                offsets[1:] = np.cumsum(counts)
                per_shard.append(offsets)
                path = os.path.join(basedir, f"{label}-{k}.r{ishard}.npy")
                self.mmaps[(k, ishard)] = np.load(path, mmap_mode="r")
            self.key_info[k] = {
                "tail_shape": metas[0]["keys"][k]["tail_shape"],
                "offsets": per_shard,
                "ndim": metas[0].get("ndims", {}).get(k, 2),
            }
        self.subset = list(subset) if subset is not None \
            else list(range(len(self.table)))
        self._cache = None
        if preload:
            self._cache = [self._fetch(i) for i in self.subset]

    def setsubset(self, subset, preload: bool = False):
        self.subset = list(subset)
        if preload:
            self._cache = [self._fetch(i) for i in self.subset]

    def _fetch(self, gid: int) -> Data:
        ishard, i = self.table[gid]
        d = Data()
        for k, info in self.key_info.items():
            offs = info["offsets"][ishard]
            lo, hi = int(offs[i]), int(offs[i + 1])
            if hi <= lo:
                continue
            arr = np.array(self.mmaps[(k, ishard)][lo:hi])
            tail = info["tail_shape"]
            if info["ndim"] == 1:
                arr = arr.reshape(hi - lo)
            else:
                arr = arr.reshape([hi - lo] + tail)
            t = torch.from_numpy(arr)
            if k == "edge_index":
                t = t.long().t().contiguous()  # [E,2] -> [2,E]
            d[k] = t
        return d

    def get_node_counts(self) -> List[int]:
        """Metadata-only per-sample node counts (used by the cost-aware
        samplers without materializing samples)."""
        key = "x" if "x" in self.key_info else "pos"
        counts = []
        for gid in self.subset:
            ishard, i = self.table[gid]
            offs = self.key_info[key]["offsets"][ishard]
            counts.append(int(offs[i + 1] - offs[i]))
        return counts

    def len(self):
        return len(self.subset)

    def get(self, idx):
        if self._cache is not None:
            return self._cache[idx]
        return self._fetch(self.subset[idx])


class DistDataset(AbstractBaseDataset):
    """DDStore-interface shim (reference distdataset.py:82-388): wraps
    a GraphStoreDataset, keeping the epoch_begin/epoch_end window API
    and get(name, buf, offset) record fetches.  Single-node: fetches
    are memmap/page-cache reads."""

    def __init__(self, dataset_or_dir, label: str = "total", comm=None,
                 ddstore_width: Optional[int] = None):
        super().__init__()
        if isinstance(dataset_or_dir, str):
            self.store = GraphStoreDataset(dataset_or_dir, label)
        else:
            self.store = dataset_or_dir
        self._in_epoch = False

    def epoch_begin(self):
        self._in_epoch = True

    def epoch_end(self):
        self._in_epoch = False

    def get_node_counts(self):
        if hasattr(self.store, "get_node_counts"):
            return self.store.get_node_counts()
        return [self.store[i].num_nodes for i in range(len(self.store))]

    def len(self):
        return len(self.store)

    def get(self, idx):
        return self.store[idx]


def allgatherv_numpy(array, comm=None):
    """Variable-length all-gather of 1-D numpy arrays over
    torch.distributed (reference distdataset.allgatherv pattern);
    returns the concatenation across ranks."""
    import numpy as np
    import torch
    import torch.distributed as dist
    if not (dist.is_available() and dist.is_initialized()):
        return np.asarray(array)
    from ..distributed import to_comm_device
    t = torch.from_numpy(np.ascontiguousarray(array))
    n, _ = to_comm_device(torch.tensor([t.numel()], dtype=torch.long))
    sizes = [torch.zeros_like(n)
             for _ in range(dist.get_world_size())]
    dist.all_gather(sizes, n)
    maxn = int(max(s.item() for s in sizes))
    pad = torch.zeros(maxn, dtype=t.dtype)
    pad[:t.numel()] = t
    pad, _ = to_comm_device(pad)
    outs = [torch.zeros_like(pad)
            for _ in range(dist.get_world_size())]
    dist.all_gather(outs, pad)
    return np.concatenate([o[:int(s.item())].cpu().numpy()
                           for o, s in zip(outs, sizes)])


class ShardedDistDataset(AbstractBaseDataset):
    """Distributed in-memory store with cross-rank sample fetches —
    the DDStore data plane (reference distdataset.py DDStore method
    'mpi'): each rank holds a shard; get(idx) for a remote sample
    fetches it from the owner inside an epoch_begin/epoch_end window.

    Transport is torch.distributed send/recv over gloo (multi-node
    capable, GPU-independent): every rank runs a service thread during
    the epoch window answering (idx) requests with the pickled sample.
    Thread-safety: requests TO rank k travel on a per-rank process
    group, so within each process the service thread and the main
    thread operate on DISJOINT groups.  Construct collectively on all
    ranks; use num_workers=0 loaders with this dataset."""

    _REQ_TAG = 7771
    _LEN_TAG = 7772
    _PAYLOAD_TAG = 7773

    def __init__(self, local_samples):
        super().__init__()
        import pickle as _pickle

        import torch.distributed as dist
        self._dist = dist
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() \
            else 1
        self._local = [_pickle.dumps(s) for s in local_samples]
        counts = [len(self._local)]
        if self.world > 1:
            from ..distributed import to_comm_device
            t, _ = to_comm_device(torch.tensor(counts,
                                               dtype=torch.long))
            sizes = [torch.zeros_like(t) for _ in range(self.world)]
            dist.all_gather(sizes, t)
            counts = [int(s) for s in sizes]
            # per-owner request/reply groups: the rank-k service thread
            # only touches groups[k]; a requesting main thread only
            # touches groups[owner != k] -> disjoint per thread
            self._req_g = [dist.new_group(backend="gloo")
                           for _ in range(self.world)]
            self._rep_g = [dist.new_group(backend="gloo")
                           for _ in range(self.world)]
        self.offsets = [0]
        for c in counts:
            self.offsets.append(self.offsets[-1] + c)
        self.total = self.offsets[-1]
        self._server = None
        self._in_epoch = False

    # -- ownership ----------------------------------------------------
    def _owner(self, idx: int) -> int:
        import bisect
        return bisect.bisect_right(self.offsets, idx) - 1

    def _serve(self):
        dist = self._dist
        g_req = self._req_g[self.rank]
        g_rep = self._rep_g[self.rank]
        req = torch.zeros(1, dtype=torch.long)
        while True:
            src = dist.recv(req, group=g_req, tag=self._REQ_TAG)
            gidx = int(req[0])
            if gidx < 0:          # stop sentinel
                return
            payload = self._local[gidx - self.offsets[self.rank]]
            buf = torch.frombuffer(bytearray(payload), dtype=torch.uint8)
            n = torch.tensor([buf.numel()], dtype=torch.long)
            dist.send(n, dst=src, group=g_rep, tag=self._LEN_TAG)
            dist.send(buf, dst=src, group=g_rep,
                      tag=self._PAYLOAD_TAG)

    def epoch_begin(self):
        if self.world > 1 and self._server is None:
            import threading
            self._server = threading.Thread(target=self._serve,
                                            daemon=True)
            self._server.start()
        self._in_epoch = True

    def epoch_end(self):
        self._in_epoch = False
        if self.world > 1 and self._server is not None:
            dist = self._dist
            dist.barrier()
            # ring stop: each rank stops its right neighbor's server
            nbr = (self.rank + 1) % self.world
            stop = torch.tensor([-1], dtype=torch.long)
            dist.send(stop, dst=nbr, group=self._req_g[nbr],
                      tag=self._REQ_TAG)
            self._server.join(timeout=60)
            self._server = None
            dist.barrier()

    # -- dataset API --------------------------------------------------
    def len(self):
        return self.total

    def get(self, idx):
        import pickle as _pickle
        owner = self._owner(idx)
        if owner == self.rank:
            return _pickle.loads(self._local[idx - self.offsets[owner]])
        if not self._in_epoch:
            raise RuntimeError(
                "remote sample fetch outside epoch_begin/epoch_end")
        dist = self._dist
        req = torch.tensor([idx], dtype=torch.long)
        dist.send(req, dst=owner, group=self._req_g[owner],
                  tag=self._REQ_TAG)
        n = torch.zeros(1, dtype=torch.long)
        dist.recv(n, src=owner, group=self._rep_g[owner],
                  tag=self._LEN_TAG)
        buf = torch.zeros(int(n[0]), dtype=torch.uint8)
        dist.recv(buf, src=owner, group=self._rep_g[owner],
                  tag=self._PAYLOAD_TAG)
        return _pickle.loads(buf.numpy().tobytes())
