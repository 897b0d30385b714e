"""FusedAdamW (flat-buffer single-kernel AdamW) vs torch.optim.AdamW:
identical trajectories on CPU (the GPU kernel implements the same
math; numerics checked in tests/test_gpu_capture.py path)."""

import torch

from hydragnn_amd.ops.fused_adamw import FusedAdamW


def _model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(torch.nn.Linear(6, 16), torch.nn.SiLU(),
                               torch.nn.Linear(16, 3))


def test_fused_adamw_matches_torch_cpu():
    m1, m2 = _model(1), _model(1)
    m2.load_state_dict(m1.state_dict())
    o1 = torch.optim.AdamW(m1.parameters(), lr=3e-3, weight_decay=0.01)
    o2 = FusedAdamW(m2.parameters(), lr=3e-3, weight_decay=0.01)
    x = torch.randn(32, 6)
    y = torch.randn(32, 3)
    for i in range(10):
        o1.zero_grad()
        torch.nn.functional.mse_loss(m1(x), y).backward()
        o1.step()
        o2.zero_grad()
        torch.nn.functional.mse_loss(m2(x), y).backward()
        o2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, rtol=1e-5, atol=1e-6), \
            (p1 - p2).abs().max()


def test_fused_adamw_params_are_flat_views():
    m = _model(2)
    o = FusedAdamW(m.parameters(), lr=1e-3)
    base = o.flat_param
    for p in m.parameters():
        assert p.data.data_ptr() >= base.data_ptr()
        assert p.grad is not None and \
            p.grad.data_ptr() >= o.flat_grad.data_ptr()
    # zero_grad keeps the views alive
    o.zero_grad()
    for p in m.parameters():
        assert p.grad is not None and float(p.grad.abs().sum()) == 0.0


def test_fused_adamw_state_roundtrip():
    m = _model(3)
    o = FusedAdamW(m.parameters(), lr=1e-3)
    x = torch.randn(8, 6)
    (m(x).square().sum()).backward()
    o.step()
    sd = o.state_dict()
    m2 = _model(3)
    o2 = FusedAdamW(m2.parameters(), lr=1e-3)
    o2.load_state_dict({"param_groups": sd["param_groups"],
                        "flat": {k: (v.clone() if v is not None else None)
                                 for k, v in sd["flat"].items()}})
    assert float(o2.step_t) == 1.0


def test_fused_adamw_bf16_master_mode_cpu():
    """bf16 params + fp32 master: trajectory tracks an fp32
    torch.optim.AdamW run within bf16 resolution."""
    m_ref = _model(4)
    m_bf = _model(4)
    m_bf.load_state_dict(m_ref.state_dict())
    m_bf = m_bf.to(torch.bfloat16)
    o_ref = torch.optim.AdamW(m_ref.parameters(), lr=3e-3,
                              weight_decay=0.01)
    o_bf = FusedAdamW(m_bf.parameters(), lr=3e-3, weight_decay=0.01)
    assert o_bf.master is not None
    x = torch.randn(32, 6)
    y = torch.randn(32, 3)
    for _ in range(8):
        o_ref.zero_grad()
        torch.nn.functional.mse_loss(m_ref(x), y).backward()
        o_ref.step()
        o_bf.zero_grad()
        torch.nn.functional.mse_loss(
            m_bf(x.bfloat16()).float(), y).backward()
        o_bf.step()
    for p1, p2 in zip(m_ref.parameters(), m_bf.parameters()):
        assert torch.allclose(p1, p2.float(), rtol=0.1, atol=0.05), \
            (p1 - p2.float()).abs().max()
