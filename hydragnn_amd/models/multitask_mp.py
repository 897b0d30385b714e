"""Branch/task model parallelism (MultiTaskModelMP).

Behavioral parity with /root/reference/hydragnn/models/
MultiTaskModelMP.py:35-542: the model is split into a shared ENCODER
replicated/sharded over the WORLD group and a per-branch DECODER owned
by a branch sub-process-group; other branches' heads are pruned from
each rank's decoder; DualOptimizer steps both halves.  Gradient sync:
encoder all-reduce over WORLD (DDP or flattened manual average),
decoder over the branch group — RCCL over xGMI on MI355X.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
from torch import nn


def average_gradients(model: nn.Module, group=None) -> None:
    """Flat-bucket gradient all-reduce (the reference averages per
    tensor, MultiTaskModelMP.py:35 — we flatten into one bucket per
    dtype, sized for the per-link-bound xGMI ring)."""
    if not dist.is_initialized():
        return
    world = dist.get_world_size(group)
    if world <= 1:
        return
    grads_by_dtype = {}
    for p in model.parameters():
        if p.grad is not None:
            grads_by_dtype.setdefault(p.grad.dtype, []).append(p.grad)
    for grads in grads_by_dtype.values():
        flat = torch.cat([g.flatten() for g in grads])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=group)
        flat /= world
        off = 0
        for g in grads:
            g.copy_(flat[off:off + g.numel()].view_as(g))
            off += g.numel()


class EncoderModel(nn.Module):
    """Shared encoder half: embedding + convs + pooling."""

    def __init__(self, base_model: nn.Module):
        super().__init__()
        self.embed_modules = nn.ModuleDict()
        self.base = base_model
        # strip decoder parameters so DDP over WORLD syncs encoder only
        self.base.graph_shared = nn.ModuleDict({})
        self.base.heads_NN = nn.ModuleList()

    def forward(self, data):
        return self.base.encoder_forward(data)


class DecoderModel(nn.Module):
    """Per-branch decoder: shared dense layers + heads of ONE branch
    (other branches' modules deleted, reference :310-343)."""

    def __init__(self, base_model: nn.Module, branch_id: int):
        super().__init__()
        self.branch_id = branch_id
        self.branch_key = f"branch-{branch_id}"
        self.head_dims = base_model.head_dims
        self.head_type = base_model.head_type
        self.num_heads = base_model.num_heads
        self.var_output = base_model.var_output
        self.config_heads = base_model.config_heads
        self.loss_weights = base_model.loss_weights
        self.loss_function = base_model.loss_function
        self.graph_shared = nn.ModuleDict({
            k: v for k, v in base_model.graph_shared.items()
            if k == self.branch_key})
        self.heads_NN = nn.ModuleList()
        for head in base_model.heads_NN:
            kept = nn.ModuleDict({
                k: v for k, v in head.items() if k == self.branch_key})
            self.heads_NN.append(kept)

    def forward(self, data, encoded):
        from .base import Base
        x, x_graph, equiv_node_feat, conv_args = encoded
        # the decode path looks the branch up by dataset_name id
        n_graphs = x_graph.shape[0]
        data["dataset_name"] = torch.full(
            (n_graphs, 1), self.branch_id, dtype=torch.long,
            device=x.device)
        return Base._decode(self, x, x_graph, equiv_node_feat, data,
                            conv_args)

    # _decode references these from Base; provide them
    @property
    def num_branches(self):
        return 2  # force branch-key lookup path (num_branches > 1)

    def _node_head_forward(self, head_mod, node_NN_type, x, equiv, batch,
                           conv_args):
        from .base import Base
        return Base._node_head_forward(self, head_mod, node_NN_type, x,
                                       equiv, batch, conv_args)

    def loss(self, pred, value, head_index):
        from .base import Base
        return Base.loss(self, pred, value, head_index)

    def loss_hpweighted(self, pred, value, head_index, var=None):
        from .base import Base
        return Base.loss_hpweighted(self, pred, value, head_index, var)


class DualOptimizer:
    """Two optimizers (encoder over WORLD, decoder over branch group)
    stepped together (reference :503)."""

    def __init__(self, optimizer1, optimizer2):
        self.optimizer1 = optimizer1
        self.optimizer2 = optimizer2

    def zero_grad(self, set_to_none: bool = True):
        self.optimizer1.zero_grad(set_to_none=set_to_none)
        self.optimizer2.zero_grad(set_to_none=set_to_none)

    def step(self):
        self.optimizer1.step()
        self.optimizer2.step()

    def state_dict(self):
        return {"optimizer1": self.optimizer1.state_dict(),
                "optimizer2": self.optimizer2.state_dict()}

    def load_state_dict(self, state):
        self.optimizer1.load_state_dict(state["optimizer1"])
        self.optimizer2.load_state_dict(state["optimizer2"])


class MultiTaskModelMP(nn.Module):
    """model = encoder (WORLD-synced) + decoder (branch-synced)."""

    def __init__(self, base_model: nn.Module, branch_id: int,
                 branch_group=None, use_ddp: bool = True):
        super().__init__()
        self.branch_id = branch_id
        self.branch_group = branch_group
        decoder = DecoderModel(base_model, branch_id)
        encoder = EncoderModel(base_model)  # strips heads in-place
        self.loss_weights = decoder.loss_weights
        self.head_type = decoder.head_type
        self.num_heads = decoder.num_heads
        self.var_output = decoder.var_output

        if use_ddp and dist.is_initialized() and \
                dist.get_world_size() > 1:
            dev = None
            if torch.cuda.is_available():
                dev = [torch.cuda.current_device()]
            encoder = nn.parallel.DistributedDataParallel(
                encoder, device_ids=dev)
            if branch_group is not None and \
                    dist.get_world_size(branch_group) > 1:
                decoder = nn.parallel.DistributedDataParallel(
                    decoder, device_ids=dev, process_group=branch_group)
        self.encoder = encoder
        self.decoder = decoder

    @property
    def module(self):
        return self

    def forward(self, data):
        encoded = self.encoder(data)
        return self.decoder(data, encoded)

    def _dec(self):
        return (self.decoder.module if hasattr(self.decoder, "module")
                else self.decoder)

    def loss(self, pred, value, head_index):
        return self._dec().loss(pred, value, head_index)

    def make_dual_optimizer(self, cls=torch.optim.AdamW, **kwargs):
        return DualOptimizer(cls(self.encoder.parameters(), **kwargs),
                             cls(self.decoder.parameters(), **kwargs))
