"""Activation / loss selection by config string.

Mirrors hydragnn.utils.model activation_function_selection /
loss_function_selection semantics (reference: hydragnn/utils/model/model.py).
"""

from __future__ import annotations

import torch
from torch import nn


def activation_function_selection(name: str) -> nn.Module:
    table = {
        "relu": nn.ReLU(),
        "selu": nn.SELU(),
        "prelu": nn.PReLU(),
        "elu": nn.ELU(),
        "lrelu_01": nn.LeakyReLU(0.1),
        "lrelu_025": nn.LeakyReLU(0.25),
        "lrelu_05": nn.LeakyReLU(0.5),
        "gelu": nn.GELU(),
        "silu": nn.SiLU(),
        "tanh": nn.Tanh(),
        "identity": nn.Identity(),
    }
    if name not in table:
        raise ValueError(f"Unknown activation function: {name}")
    return table[name]


class RMSELoss(nn.Module):
    def __init__(self):
        super().__init__()
        self.mse = nn.MSELoss()

    def forward(self, pred, target):
        return torch.sqrt(self.mse(pred, target))


def loss_function_selection(name: str) -> nn.Module:
    table = {
        "mse": nn.MSELoss(),
        "mae": nn.L1Loss(),
        "rmse": RMSELoss(),
        "smooth_l1": nn.SmoothL1Loss(),
        "huber": nn.HuberLoss(),
        "GaussianNLLLoss": nn.GaussianNLLLoss(),
    }
    if name not in table:
        raise ValueError(f"Unknown loss function: {name}")
    return table[name]
