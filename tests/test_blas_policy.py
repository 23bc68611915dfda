"""Numerics of the MI355X GEMM-policy helpers on CPU (fp32): chunked
split-K reduction and the custom IH-projection autograd must match plain
torch ops exactly (same math, different scheduling)."""
import torch

from fmda_amd.ops.blas import chunked_outer
from fmda_amd.ops.interface import _IHProjection


def test_chunked_outer_matches_matmul():
    g = torch.Generator().manual_seed(0)
    dg = torch.randn(1024, 24, generator=g)
    x = torch.randn(1024, 12, generator=g)
    ref = dg.t() @ x
    got = chunked_outer(dg, x, chunks=16)
    assert torch.allclose(got, ref, atol=1e-4)


def test_chunked_outer_odd_rows_falls_back():
    g = torch.Generator().manual_seed(1)
    dg = torch.randn(1023, 8, generator=g)  # prime-ish M: no even split
    x = torch.randn(1023, 4, generator=g)
    got = chunked_outer(dg, x, chunks=64)
    assert torch.allclose(got, dg.t() @ x, atol=1e-4)


def test_ih_projection_grads_match_addmm():
    g = torch.Generator().manual_seed(2)
    x = torch.randn(256, 12, generator=g, requires_grad=True)
    w = torch.randn(24, 12, generator=g, requires_grad=True)
    b = torch.randn(24, generator=g, requires_grad=True)
    dout = torch.randn(256, 24, generator=g)

    out1 = _IHProjection.apply(x, w, b)
    out1.backward(dout)
    g1 = (x.grad.clone(), w.grad.clone(), b.grad.clone())
    x.grad = w.grad = b.grad = None

    out2 = torch.addmm(b, x, w.t())
    out2.backward(dout)

    assert torch.allclose(out1, out2, atol=1e-5)
    for a, r in zip(g1, (x.grad, w.grad, b.grad)):
        assert torch.allclose(a, r, atol=1e-4)


def test_ih_projection_skips_dx_when_not_needed():
    g = torch.Generator().manual_seed(3)
    x = torch.randn(64, 8, generator=g)  # no grad
    w = torch.randn(12, 8, generator=g, requires_grad=True)
    b = torch.randn(12, generator=g, requires_grad=True)
    out = _IHProjection.apply(x, w, b)
    out.sum().backward()
    assert w.grad is not None and b.grad is not None
