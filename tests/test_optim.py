"""FusedClipAdam vs torch clip_grad_norm_ + Adam (same math)."""
import pytest
import torch
import torch.nn as nn

from fmda_amd.optim import FusedClipAdam


def _models(device="cpu"):
    torch.manual_seed(0)
    m1 = nn.Sequential(nn.Linear(13, 7), nn.Tanh(), nn.Linear(7, 3)).to(device)
    m2 = nn.Sequential(nn.Linear(13, 7), nn.Tanh(), nn.Linear(7, 3)).to(device)
    m2.load_state_dict(m1.state_dict())
    return m1, m2


def _run_pair(device, clip, steps=5, tol=1e-6):
    m1, m2 = _models(device)
    o1 = FusedClipAdam(m1.parameters(), lr=1e-2, clip=clip)
    o2 = torch.optim.Adam(m2.parameters(), lr=1e-2)
    g = torch.Generator().manual_seed(3)
    for i in range(steps):
        x = torch.randn(32, 13, generator=g).to(device)
        y = torch.randn(32, 3, generator=g).to(device)
        for m, o in ((m1, o1), (m2, o2)):
            o.zero_grad()
            loss = ((m(x) - y) ** 2).mean() * 40  # large grads -> clip active
            loss.backward()
            if o is o2 and clip > 0:
                nn.utils.clip_grad_norm_(m2.parameters(), clip)
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=tol), (p1 - p2).abs().max()


def test_eager_fallback_matches_torch_with_clip():
    _run_pair("cpu", clip=0.05)


def test_eager_fallback_matches_torch_no_clip():
    _run_pair("cpu", clip=0.0)


@pytest.mark.gpu
def test_fused_kernel_matches_torch_with_clip():
    _run_pair("cuda", clip=0.05, tol=1e-5)


@pytest.mark.gpu
def test_fused_kernel_matches_torch_no_clip():
    _run_pair("cuda", clip=0.0, tol=1e-5)
