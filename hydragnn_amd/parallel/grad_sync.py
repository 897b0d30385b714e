"""fp32 flat-bucket gradient synchronization for data parallelism.

MI355X-native DP gradient plane used by the hipGraph-captured train
step (``hydragnn_amd/train/captured.py``): the captured graph produces
``param.grad`` in the parameter dtype (bf16 for the headline config);
after replay this synchronizer casts all grads into one persistent
fp32 flat buffer, all-reduces it over RCCL/xGMI in buckets, and casts
the averaged result back.  fp32 accumulation over ranks matches the
reference's DDP reduction precision (reference wraps fp32 models in
torch DDP, /root/reference/hydragnn/utils/distributed/
distributed.py:489-510) without paying fp32 parameters in compute.

Bucketing: xGMI is point-to-point (7 links x ~153 GB/s); ring
all-reduce is per-link bound, so a handful of multi-MB buckets issued
back-to-back keeps the rings busy without launch overhead.  For the
MACE headline model the whole gradient is ~2 MB -> one bucket.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class FlatGradSync:
    """Persistent fp32 gradient bucket: ``sync()`` averages
    ``param.grad`` across ranks in fp32.

    Parameters must already have ``.grad`` allocated (run one backward
    first); grads must keep the same storage across steps (true for
    the captured path, which zeroes in place instead of
    ``set_to_none``).
    """

    def __init__(self, params, process_group=None,
                 bucket_cap_mb: float = 64.0):
        self.params = [p for p in params if p.requires_grad]
        self.group = process_group
        self.world = (dist.get_world_size(process_group)
                      if dist.is_initialized() else 1)
        self._views: Optional[List[torch.Tensor]] = None
        self._buckets: List[torch.Tensor] = []
        self.bucket_cap = int(bucket_cap_mb * 2 ** 20 / 4)

    def _lazy_init(self):
        grads = [p.grad for p in self.params]
        assert all(g is not None for g in grads), \
            "FlatGradSync: run a backward pass before sync()"
        total = sum(g.numel() for g in grads)
        device = grads[0].device
        flat = torch.zeros(total, dtype=torch.float32, device=device)
        views = []
        off = 0
        for g in grads:
            views.append(flat[off:off + g.numel()].view_as(g))
            off += g.numel()
        self._flat = flat
        self._views = views
        self._grads = grads
        self._buckets = list(flat.split(self.bucket_cap)) \
            if total > self.bucket_cap else [flat]

    def sync(self):
        """Average grads across ranks (fp32 accumulate).  No-op at
        world size 1 (no cast round-trip either)."""
        if self.world <= 1:
            return
        if self._views is None:
            self._lazy_init()
        # cast grads -> fp32 flat buffer
        torch._foreach_copy_(self._views, self._grads)
        self._flat.div_(self.world)
        handles = [dist.all_reduce(b, group=self.group, async_op=True)
                   for b in self._buckets]
        for h in handles:
            h.wait()
        # cast back so the optimizer sees synchronized grads
        torch._foreach_copy_(self._grads, self._views)

    __call__ = sync
