"""GPU numerics tests: HIP kernels vs the plain PyTorch fp32 reference.

Every test is @pytest.mark.gpu and requires the in-tree _fmda_hip extension
(which fails loudly if missing on a GPU box — fmda_amd/ops/__init__.py).
"""
import pytest
import torch
import torch.nn as nn

pytestmark = pytest.mark.gpu


def _ext():
    from fmda_amd.ops import load_extension
    return load_extension()


def test_mfma_fragment_layout():
    """Guards the assumed v_mfma_f32_16x16x32_bf16 A/B/C fragment layout."""
    ext = _ext()
    g = torch.Generator().manual_seed(0)
    A = torch.randn(16, 32, generator=g).bfloat16().cuda()
    B = torch.randn(32, 16, generator=g).bfloat16().cuda()
    C = ext.mfma_selftest(A, B)
    ref = (A.float() @ B.float())
    assert torch.allclose(C, ref, atol=2e-2, rtol=2e-2), (C - ref).abs().max()


@pytest.mark.parametrize("H,T,B,n_dir", [
    (16, 7, 5, 2), (32, 12, 33, 2), (64, 30, 8, 2), (128, 20, 32, 2),
    (128, 120, 40, 2), (32, 9, 4, 1), (256, 10, 16, 2), (512, 6, 16, 2),
])
def test_gru_fwd_fp32_matches_reference(H, T, B, n_dir):
    """fp32 kernel path (VALU oracle) vs the golden torch GRU, tight tol."""
    ext = _ext()
    torch.manual_seed(1)
    gi = torch.randn(B, T, n_dir * 3 * H).cuda()
    w = torch.randn(n_dir, 3 * H, H).cuda() * 0.3
    bhh = torch.randn(n_dir, 3 * H).cuda() * 0.1
    out, hlast = ext.gru_fwd(gi, w, bhh)
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh)
    # fp32 summation-order roundoff: ~1e-5 per long dot product, amplified
    # through the recurrence for long T (measured 2.4e-3 at T=120 with the
    # same-tolerance torch reference; backward parity holds at 1e-4 rel).
    tol = 5e-3 if T >= 100 else 1e-4
    assert torch.allclose(out, out_ref, atol=tol), (out - out_ref).abs().max()
    assert torch.allclose(hlast, hlast_ref, atol=tol)


@pytest.mark.parametrize("H,T,B", [(32, 12, 33), (128, 60, 32), (512, 8, 16)])
def test_gru_fwd_bf16_close_to_fp32_reference(H, T, B):
    ext = _ext()
    torch.manual_seed(2)
    n_dir = 2
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh)
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh)
    err = (out.float() - out_ref).abs().max()
    assert err < 0.05, err  # bf16 GEMM + bf16 h shadow, fp32 state
    assert (hlast - hlast_ref).abs().max() < 0.05


def test_gru_fwd_batch1_lds_resident_kernel():
    """B=1 routes to the LDS-resident streaming kernel (gru_fwd_b1_kernel);
    it must match the fp32 reference AND the batch-tiled kernel's own
    result at B=1 (same bf16 rounding chain)."""
    ext = _ext()
    torch.manual_seed(5)
    H, T, n_dir = 128, 120, 2
    gi = (torch.randn(1, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh)
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh)
    assert (out.float() - out_ref).abs().max() < 0.05
    assert (hlast - hlast_ref).abs().max() < 0.05
    # batch-tiled kernel on the same inputs padded to B=2 (forces v3 path)
    gi2 = torch.cat([gi, gi], dim=0)
    out2, hlast2 = ext.gru_fwd(gi2.bfloat16(), w.bfloat16(), bhh)
    assert (out.float() - out2[:1].float()).abs().max() < 0.02


def test_gru_fwd_batch1_unidirectional():
    ext = _ext()
    torch.manual_seed(6)
    H, T = 128, 64
    gi = (torch.randn(1, T, 3 * H) * 0.5).cuda()
    w = (torch.randn(1, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(1, 3 * H) * 0.1).cuda()
    out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh)
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh)
    assert (out.float() - out_ref).abs().max() < 0.05
    assert (hlast - hlast_ref).abs().max() < 0.05


def test_gru_fwd_batch1_long_sequence_fallback():
    """T past the LDS-residency cap must fall back to the batch-tiled
    kernel with identical semantics."""
    ext = _ext()
    torch.manual_seed(9)
    H, T, n_dir = 128, 256, 2   # 256*(3*128+8)*2 B > 150 KB -> v3 path
    gi = (torch.randn(1, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh)
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh)
    assert (out.float() - out_ref).abs().max() < 0.06
    assert (hlast - hlast_ref).abs().max() < 0.06


def _gru_ref_from_gi(gi, w, bhh, h0=None):
    """Golden recurrence on precomputed input projections (fp32 torch)."""
    B, T, _ = gi.shape
    n_dir, threeH, H = w.shape
    out = torch.zeros(B, T, n_dir * H, device=gi.device)
    hlast = torch.zeros(n_dir, B, H, device=gi.device)
    for d in range(n_dir):
        h = (h0[d].clone() if h0 is not None
             else torch.zeros(B, H, device=gi.device))
        steps = range(T - 1, -1, -1) if d == 1 else range(T)
        for t in steps:
            g = gi[:, t, d * threeH:(d + 1) * threeH]
            gh = h @ w[d].t() + bhh[d]
            i_r, i_z, i_n = g.chunk(3, -1)
            h_r, h_z, h_n = gh.chunk(3, -1)
            r = torch.sigmoid(i_r + h_r)
            z = torch.sigmoid(i_z + h_z)
            n = torch.tanh(i_n + r * h_n)
            h = (1 - z) * n + z * h
            out[:, t, d * H:(d + 1) * H] = h
        hlast[d] = h
    return out, hlast


@pytest.mark.parametrize("H,T,B,n_dir", [
    (16, 6, 5, 2), (32, 10, 33, 2), (64, 16, 8, 2), (128, 25, 32, 2),
    (256, 8, 16, 2), (32, 7, 4, 1),
])
def test_gru_backward_fp32_matches_autograd(H, T, B, n_dir):
    """Full BPTT: kernel-backed autograd vs torch autograd on the golden
    recurrence, fp32, tight tolerance."""
    from fmda_amd.ops.interface import gru_directions
    torch.manual_seed(3)
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    dO = torch.randn(B, T, n_dir * H).cuda()
    dH = torch.randn(n_dir, B, H).cuda()

    gi1 = gi.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    b1 = bhh.clone().requires_grad_(True)
    out1, h1 = gru_directions(gi1, w1, b1)
    (out1 * dO).sum().add_((h1 * dH).sum()).backward()

    gi2 = gi.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = bhh.clone().requires_grad_(True)
    out2, h2 = _gru_ref_from_gi_autograd(gi2, w2, b2)
    (out2 * dO).sum().add_((h2 * dH).sum()).backward()

    assert torch.allclose(out1, out2, atol=1e-5)
    for a, b in [(gi1.grad, gi2.grad), (w1.grad, w2.grad), (b1.grad, b2.grad)]:
        scale = b.abs().max().clamp(min=1.0)
        assert ((a - b).abs().max() / scale) < 1e-4, (a - b).abs().max()


def _gru_ref_from_gi_autograd(gi, w, bhh, h0=None):
    B, T, _ = gi.shape
    n_dir, threeH, H = w.shape
    outs = []
    hlast = []
    for d in range(n_dir):
        h = h0[d] if h0 is not None else torch.zeros(B, H, device=gi.device)
        hs = [None] * T
        steps = range(T - 1, -1, -1) if d == 1 else range(T)
        for t in steps:
            g = gi[:, t, d * threeH:(d + 1) * threeH]
            gh = h @ w[d].t() + bhh[d]
            i_r, i_z, i_n = g.chunk(3, -1)
            h_r, h_z, h_n = gh.chunk(3, -1)
            r = torch.sigmoid(i_r + h_r)
            z = torch.sigmoid(i_z + h_z)
            n = torch.tanh(i_n + r * h_n)
            h = (1 - z) * n + z * h
            hs[t] = h
        outs.append(torch.stack(hs, 1))
        hlast.append(h)
    return torch.cat(outs, -1), torch.stack(hlast, 0)


def test_bigru_model_gpu_matches_cpu():
    """Whole-model forward on GPU (HIP engine) vs CPU (ATen), fp32."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(7)
    m = BiGRU(32, 24, 4, n_layers=2, spatial_dropout=False, dropout=0.0)
    m.eval()
    x = torch.randn(6, 15, 24)
    with torch.no_grad():
        ref = m(x)
        got = m.cuda()(x.cuda()).cpu()
    assert torch.allclose(ref, got, atol=5e-4), (ref - got).abs().max()


def test_model_train_step_gpu_bf16():
    """One full fwd+bwd+step on GPU in bf16 runs and updates weights."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(8)
    m = BiGRU(128, 96, 4, n_layers=2, spatial_dropout=False,
              dropout=0.1).cuda()
    opt = torch.optim.Adam(m.parameters(), lr=1e-3)
    x = torch.randn(32, 20, 96, device="cuda", dtype=torch.bfloat16)
    y = (torch.rand(32, 4, device="cuda") < 0.3).float()
    before = m.linear.weight.detach().clone()
    logits = m(x)
    loss = nn.functional.binary_cross_entropy_with_logits(logits.float(), y)
    loss.backward()
    nn.utils.clip_grad_norm_(m.parameters(), 50.0)
    opt.step()
    assert not torch.equal(before, m.linear.weight.detach())
    assert torch.isfinite(loss).item()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("B", [1, 33, 700])
def test_dirsum_pool_matches_eager(dtype, B):
    """Fused pooling kernel vs the plain torch ops (fwd + bwd). B=1/33 hit
    the wave-per-pair small-grid variant, B=700 the thread-per-pair one."""
    from fmda_amd.ops.interface import dirsum_pool
    torch.manual_seed(7)
    T, H = 19, 24
    out = (torch.randn(B, T, 2 * H) * 2).to(dtype).cuda().requires_grad_(True)
    mx, av = dirsum_pool(out, 2)
    gm = torch.randn(B, H).to(dtype).cuda()
    ga = torch.randn(B, H).to(dtype).cuda()
    ((mx * gm).sum() + (av * ga).sum()).backward()
    g1 = out.grad.clone()
    out.grad = None

    s = out[:, :, :H].float() + out[:, :, H:].float()
    mx2 = s.max(dim=1).values
    av2 = s.sum(dim=1) / T
    ((mx2 * gm.float()).sum() + (av2 * ga.float()).sum()).backward()
    tol = 1e-5 if dtype == torch.float32 else 5e-2
    assert torch.allclose(mx.float(), mx2.to(mx.dtype).float(), atol=tol)
    assert torch.allclose(av.float(), av2.to(av.dtype).float(), atol=tol)
    assert torch.allclose(g1.float(), out.grad.float(), atol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_head_loss_matches_eager(dtype):
    """Fused head GEMM + BCEWithLogitsLoss(weight, pos_weight) vs the plain
    torch ops: loss value and x/W/b gradients."""
    from fmda_amd.ops.interface import fused_head_loss
    torch.manual_seed(11)
    B, K, C = 37, 96, 4
    x = (torch.randn(B, K) * 0.5).to(dtype).cuda().requires_grad_(True)
    W = torch.randn(C, K, device="cuda") * 0.2
    bparm = torch.randn(C, device="cuda") * 0.1
    W.requires_grad_(True); bparm.requires_grad_(True)
    y = (torch.rand(B, C, device="cuda") < 0.3).float()
    wgt = torch.rand(C, device="cuda") * 3 + 0.5
    pw = torch.rand(C, device="cuda") * 4 + 0.5

    loss1, logits1 = fused_head_loss(x, W, bparm, y, wgt, pw)
    loss1.backward()
    g1 = (x.grad.clone(), W.grad.clone(), bparm.grad.clone())
    x.grad = W.grad = bparm.grad = None

    lf = nn.BCEWithLogitsLoss(weight=wgt, pos_weight=pw)
    logits2 = torch.nn.functional.linear(x.float(), W.to(dtype).float(),
                                         bparm.to(dtype).float())
    loss2 = lf(logits2, y)
    loss2.backward()

    tol = 1e-5 if dtype == torch.float32 else 3e-2
    assert torch.allclose(loss1.float(), loss2, atol=tol, rtol=tol)
    assert torch.allclose(logits1, logits2, atol=5e-2)
    rel = lambda a, b: ((a.float() - b.float()).abs().max() /
                        b.float().abs().max().clamp(min=1e-3))
    assert rel(g1[0], x.grad) < (1e-4 if dtype == torch.float32 else 5e-2)
    assert rel(g1[1], W.grad) < (1e-4 if dtype == torch.float32 else 5e-2)
    assert rel(g1[2], bparm.grad) < (1e-4 if dtype == torch.float32 else 5e-2)


@pytest.mark.parametrize("B,H", [(64, 128), (33, 128), (48, 512), (300, 512),
                                 (33, 64), (16, 256), (24, 32)])
def test_bigru_layer_bf16_all_grads_vs_fp32_reference(B, H):
    """Full single-layer node (pack + projection + recurrence) in bf16 vs
    an fp32 autograd reference: checks out/h_last AND every parameter
    gradient (w_ih, w_hh, b_ih, b_hh, both directions) — the bias grads
    come from in-kernel register accumulators and are otherwise untested."""
    from fmda_amd.ops.interface import _BiGRULayer
    T, F = 11, 96
    torch.manual_seed(5)
    params = []
    for d in range(2):
        params += [
            (torch.randn(3 * H, F, device="cuda") * 0.2).requires_grad_(True),
            (torch.randn(3 * H, H, device="cuda") * 0.2).requires_grad_(True),
            (torch.randn(3 * H, device="cuda") * 0.1).requires_grad_(True),
            (torch.randn(3 * H, device="cuda") * 0.1).requires_grad_(True),
        ]
    x = (torch.randn(B, T, F, device="cuda") * 0.5).bfloat16()
    dO = torch.randn(B, T, 2 * H, device="cuda")
    dH = torch.randn(2, B, H, device="cuda")

    out, hl = _BiGRULayer.apply(x, H, *params)
    ((out.float() * dO).sum() + (hl * dH).sum()).backward()
    got = [p.grad.clone() for p in params]
    for p in params:
        p.grad = None

    # fp32 reference through plain autograd
    x32 = x.float()
    gi = torch.cat([x32.reshape(-1, F) @ params[0].t() + params[2],
                    x32.reshape(-1, F) @ params[4].t() + params[6]],
                   dim=-1).view(B, T, -1)
    w = torch.stack([params[1], params[5]], dim=0)
    bhh = torch.stack([params[3], params[7]], dim=0)
    out2, hl2 = _gru_ref_from_gi_autograd(gi, w, bhh)
    ((out2 * dO).sum() + (hl2 * dH).sum()).backward()

    # max-error for H=128; at H=512 the bf16 rounding amplifies through
    # the recurrence (per-step gain > 1 at this K), so use relative-L2
    # there — a protocol bug (stale/missing exchange slice) produces O(1)
    # banded errors that these bounds still catch.
    if H <= 128:
        assert (out.float() - out2).abs().max() < 0.12, "fwd mismatch"
    else:
        rl2 = (out.float() - out2).norm() / out2.norm()
        assert rl2 < 0.03, f"fwd rel-L2 {rl2}"
    names = ["w_ih0", "w_hh0", "b_ih0", "b_hh0",
             "w_ih1", "w_hh1", "b_ih1", "b_hh1"]
    for nm, a, p in zip(names, got, params):
        ref = p.grad
        if H <= 128:
            rel = (a - ref).abs().max() / ref.abs().max().clamp(min=1e-2)
            assert rel < 6e-2, f"{nm}: rel={rel}"
        else:
            rel = (a - ref).norm() / ref.norm().clamp(min=1e-2)
            assert rel < 8e-2, f"{nm}: rel-L2={rel}"


def test_training_step_reproducible():
    """Two identical training steps from identical state produce the same
    parameters within atomic-accumulation jitter (SURVEY.md section 4 (d);
    db_hh/db_ih use atomicAdd, so bitwise equality is not expected)."""
    from fmda_amd.optim import FusedClipAdam
    results = []
    for _ in range(2):
        torch.manual_seed(77)
        from fmda_amd.models import BiGRU
        m = BiGRU(128, 96, 4, n_layers=2, spatial_dropout=False,
                  dropout=0.0).cuda()
        opt = FusedClipAdam(m.parameters(), lr=1e-3, clip=50.0)
        g = torch.Generator().manual_seed(5)
        x = torch.randn(64, 30, 96, generator=g).bfloat16().cuda()
        y = (torch.rand(64, 4, generator=g) < 0.3).float().cuda()
        for _ in range(3):
            opt.zero_grad(set_to_none=False)
            logits = m(x)
            loss = nn.functional.binary_cross_entropy_with_logits(
                logits.float(), y)
            loss.backward()
            opt.step()
        results.append([p.detach().clone() for p in m.parameters()])
    for a, b in zip(*results):
        assert torch.allclose(a, b, atol=1e-5, rtol=1e-4), \
            (a - b).abs().max()


def test_reference_train_api_on_gpu():
    """The reference train_model/evaluate_model epoch loops run on the HIP
    engine (biGRU_model.py:162-286 semantics on cuda)."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(3)
    m = BiGRU(32, 16, 4, n_layers=1, spatial_dropout=False,
              dropout=0.2).cuda()
    m.add_loss_fn(nn.BCEWithLogitsLoss())
    m.add_optimizer(torch.optim.Adam(m.parameters(), lr=1e-3))
    m.add_device(torch.device("cuda"))
    g = torch.Generator().manual_seed(9)
    batches = [(torch.randn(8, 12, 16, generator=g),
                (torch.rand(8, 1, 4, generator=g) < 0.3).float())
               for _ in range(4)]
    acc, ham, loss, fbeta = m.train_model(iter(batches))
    assert 0.0 <= acc <= 1.0 and 0.0 <= ham <= 1.0
    acc2, ham2, fb2, pred_tot, tgt_tot = m.evaluate_model(iter(batches))
    assert pred_tot.shape == (32, 4) and tgt_tot.shape == (32, 4)


def test_unidirectional_model_on_gpu():
    """bidirectional=False through the HIP engine (n_dir=1 grid) vs the
    CPU golden path."""
    torch.manual_seed(21)
    from fmda_amd.models import BiGRU
    m = BiGRU(128, 24, 4, n_layers=1, spatial_dropout=False, dropout=0.0,
              bidirectional=False)
    x = torch.randn(9, 14, 24)
    m.eval()
    ref = m(x)
    got = m.cuda()(x.cuda()).cpu()
    assert torch.allclose(got, ref, atol=5e-3), (got - ref).abs().max()


@pytest.mark.parametrize("B,T,F,H", [(48, 24, 64, 128), (16, 8, 64, 512)])
def test_deferred_dropout_backward_matches_explicit(B, T, F, H):
    """The dropout mask applied inside gru_bwd's d_out read (deferred
    path) must produce the same layer gradients as the explicit
    _FusedDropout backward pass (same seed, same counter-based mask) —
    on the v3 (H=128) and column-split (H=512) kernels."""
    from fmda_amd.ops.interface import _BiGRULayer, _FusedDropout
    torch.manual_seed(3)
    p_drop, seed = 0.3, 987654321
    masters = []
    for _ in range(2):
        masters += [torch.randn(3 * H, F, device="cuda") * 0.2,
                    torch.randn(3 * H, H, device="cuda") * 0.2,
                    torch.randn(3 * H, device="cuda") * 0.1,
                    torch.randn(3 * H, device="cuda") * 0.1]
    x = (torch.randn(B, T, F, device="cuda") * 0.5).bfloat16()
    gup = torch.randn(B, T, 2 * H, device="cuda").bfloat16()
    ghl = torch.randn(2, B, H, device="cuda")

    def run(deferred):
        ms = [m.detach().clone().requires_grad_(True) for m in masters]
        if deferred:
            # fully fused: fwd kernel emits the dropped copy, BPTT kernel
            # recomputes the mask at its d_out read
            out, hl, dropped = _BiGRULayer.apply(x, H, *ms[:4], *ms[4:],
                                                 p_drop, seed)
        else:
            out, hl = _BiGRULayer.apply(x, H, *ms[:4], *ms[4:], 0.0, 0)
            dropped = _FusedDropout.apply(out, p_drop, seed)
        ((dropped.float() * gup.float()).sum()
         + (hl * ghl).sum()).backward()
        return [m.grad.clone() for m in ms], dropped.detach()

    ga, da = run(True)
    gb, db = run(False)
    assert torch.equal(da, db)  # forward mask identical by construction
    tol = 2e-2 if H <= 128 else 8e-2   # bf16 compounding at H=512
    for i, (a, b) in enumerate(zip(ga, gb)):
        rel = (a - b).norm() / b.norm().clamp(min=1e-12)
        assert rel < tol, (i, float(rel))


def test_padded_hidden_h8_model_matches_cpu():
    """The real reference checkpoint is H=8 -> padded to Hp=16 inside the
    engine; the whole padding path (gate-row/col zero-pad, unpad slices)
    must match the CPU ATen forward AND produce matching gradients."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(11)
    m = BiGRU(8, 108, 4, n_layers=1, spatial_dropout=False, dropout=0.0)
    m.eval()
    x = torch.randn(5, 12, 108)
    with torch.no_grad():
        ref = m(x)
        got = m.cuda()(x.cuda()).cpu()
    assert torch.allclose(ref, got, atol=5e-4), (ref - got).abs().max()

    # gradient parity through the padded path (fp32 kernels)
    m.train()
    mc = BiGRU(8, 108, 4, n_layers=1, spatial_dropout=False, dropout=0.0)
    mc.load_state_dict({k: v.cpu() for k, v in m.state_dict().items()})
    y = (torch.rand(5, 4) < 0.4).float()
    for mod, dev in ((m, "cuda"), (mc, "cpu")):
        mod.train()
        out = mod(x.to(dev))
        torch.nn.functional.binary_cross_entropy_with_logits(
            out.float(), y.to(dev)).backward()
    for (n1, p1), (n2, p2) in zip(m.named_parameters(),
                                  mc.named_parameters()):
        assert torch.allclose(p1.grad.cpu(), p2.grad, atol=2e-4), n1


def test_padded_hidden_h100_bf16_train_step():
    """H=100 -> Hp=128 on the bf16 v3 kernels: full train step through the
    reference API runs and matches the fp32 CPU gradients loosely."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(12)
    m = BiGRU(100, 96, 4, n_layers=2, spatial_dropout=False,
              dropout=0.0).cuda()
    x = torch.randn(16, 20, 96, device="cuda", dtype=torch.bfloat16)
    y = (torch.rand(16, 4, device="cuda") < 0.3).float()
    out = m(x)
    assert out.shape == (16, 4)
    loss = torch.nn.functional.binary_cross_entropy_with_logits(
        out.float(), y)
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_fused_dropout_statistics_and_backward():
    """Counter-based dropout: keep-rate ~ 1-p, kept elements scaled by
    1/(1-p), and backward applies the IDENTICAL mask (recomputed)."""
    from fmda_amd.ops.interface import _FusedDropout
    torch.manual_seed(1)
    x = torch.ones(4096, 257, device="cuda").bfloat16().requires_grad_(True)
    p = 0.4
    y = _FusedDropout.apply(x, p, 12345)
    kept = (y != 0)
    rate = kept.float().mean().item()
    assert abs(rate - (1 - p)) < 0.01, rate
    scale = y.float()[kept].mean().item()
    assert abs(scale - 1 / (1 - p)) < 0.02, scale
    g = torch.ones_like(y)
    y.backward(g)
    # backward mask identical to forward mask
    assert torch.equal((x.grad != 0), kept)


# ---------------------------------------------------------------------------
# Initial hidden state (h0) support — nn.GRU signature parity
# (reference biGRU_model.py:102 `self.gru(input_seq, hidden)`).
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("H,T,B,n_dir", [
    (16, 7, 5, 2), (32, 12, 33, 2), (128, 20, 32, 2), (64, 9, 4, 1),
    (256, 8, 16, 2), (512, 6, 16, 2),
])
def test_gru_fwd_h0_fp32_matches_reference(H, T, B, n_dir):
    ext = _ext()
    torch.manual_seed(11)
    gi = torch.randn(B, T, n_dir * 3 * H).cuda()
    w = torch.randn(n_dir, 3 * H, H).cuda() * 0.3
    bhh = torch.randn(n_dir, 3 * H).cuda() * 0.1
    h0 = torch.randn(n_dir, B, H).cuda() * 0.5
    out, hlast = ext.gru_fwd(gi, w, bhh, h0.contiguous())
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh, h0)
    assert torch.allclose(out, out_ref, atol=1e-4), (out - out_ref).abs().max()
    assert torch.allclose(hlast, hlast_ref, atol=1e-4)


@pytest.mark.parametrize("H,T,B", [(128, 20, 32), (32, 9, 7), (512, 5, 8)])
def test_gru_fwd_h0_bf16(H, T, B):
    """bf16 paths with h0: v3 (H=128), v2 (H=32), and the batch-parallel
    fallback the Hp=512 h0 call routes to."""
    ext = _ext()
    torch.manual_seed(12)
    n_dir = 2
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    h0 = (torch.randn(n_dir, B, H) * 0.5).cuda()
    out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh, h0.contiguous())
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh, h0)
    assert (out.float() - out_ref).abs().max() < 0.05
    assert (hlast - hlast_ref).abs().max() < 0.05


def test_gru_fwd_h0_batch1_kernel():
    """The LDS-resident batch-1 streaming kernel honors h0."""
    ext = _ext()
    torch.manual_seed(13)
    H, T, n_dir = 128, 40, 2
    gi = (torch.randn(1, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    h0 = (torch.randn(n_dir, 1, H) * 0.5).cuda()
    out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh, h0.contiguous())
    out_ref, hlast_ref = _gru_ref_from_gi(gi, w, bhh, h0)
    assert (out.float() - out_ref).abs().max() < 0.05
    assert (hlast - hlast_ref).abs().max() < 0.05


@pytest.mark.parametrize("H,T,B,n_dir", [
    (32, 10, 33, 2), (128, 25, 32, 2), (64, 7, 4, 1),
])
def test_gru_backward_h0_fp32_matches_autograd(H, T, B, n_dir):
    """BPTT with a real initial hidden state: dgi/dW/db AND dh0 against
    torch autograd (the t=0 dGh (x) h0 term of dW_hh is the tricky bit)."""
    from fmda_amd.ops.interface import gru_directions
    torch.manual_seed(14)
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    h0 = (torch.randn(n_dir, B, H) * 0.5).cuda()
    dO = torch.randn(B, T, n_dir * H).cuda()
    dH = torch.randn(n_dir, B, H).cuda()

    gi1 = gi.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    b1 = bhh.clone().requires_grad_(True)
    h01 = h0.clone().requires_grad_(True)
    out1, hl1 = gru_directions(gi1, w1, b1, h01)
    (out1 * dO).sum().add_((hl1 * dH).sum()).backward()

    gi2 = gi.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = bhh.clone().requires_grad_(True)
    h02 = h0.clone().requires_grad_(True)
    out2, hl2 = _gru_ref_from_gi_autograd(gi2, w2, b2, h02)
    (out2 * dO).sum().add_((hl2 * dH).sum()).backward()

    assert torch.allclose(out1, out2, atol=1e-5)
    for nm, a, b in [("dgi", gi1.grad, gi2.grad), ("dw", w1.grad, w2.grad),
                     ("dbhh", b1.grad, b2.grad), ("dh0", h01.grad, h02.grad)]:
        scale = b.abs().max().clamp(min=1.0)
        assert ((a - b).abs().max() / scale) < 1e-4, (nm, (a - b).abs().max())


def test_bigru_model_h0_gpu_matches_cpu():
    """Model-level nn.GRU-signature parity: forward(x, hidden) with a
    nonzero hidden on the HIP engine vs the CPU ATen oracle, plus gradient
    flow back to the hidden tensor."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(15)
    m = BiGRU(32, 24, 4, n_layers=2, spatial_dropout=False, dropout=0.0)
    x = torch.randn(6, 15, 24)
    h0 = torch.randn(2 * 2, 6, 32) * 0.5

    h0_cpu = h0.clone().requires_grad_(True)
    logits_cpu = m(x, h0_cpu)
    logits_cpu.sum().backward()

    mg = BiGRU(32, 24, 4, n_layers=2, spatial_dropout=False, dropout=0.0)
    mg.load_state_dict(m.state_dict())
    mg = mg.cuda()
    h0_gpu = h0.clone().cuda().requires_grad_(True)
    logits_gpu = mg(x.cuda(), h0_gpu)
    logits_gpu.sum().backward()

    assert (logits_gpu.cpu() - logits_cpu).abs().max() < 1e-3
    assert (h0_gpu.grad.cpu() - h0_cpu.grad).abs().max() < 1e-3


# ---------------------------------------------------------------------------
# Spatial (channel) dropout — reference Dropout2d semantics on the engine
# (biGRU_model.py:50-52,87-94; the reference API default spatial_dropout=True)
# ---------------------------------------------------------------------------

def test_spatial_dropout_channel_mask_and_backward():
    ext = _ext()
    B, T, F = 8, 33, 108
    p = 0.4
    x = torch.ones(B, T, F, device="cuda", dtype=torch.bfloat16)
    y = ext.spatial_dropout_fused(x, p, 1234)
    yf = y.float()
    # channel mask: for each (b, f) the value is identical across ALL t
    assert (yf == yf[:, :1, :]).all(), "mask varies along time"
    # kept channels carry exactly the 1/(1-p) scale
    kept = yf[:, 0, :][yf[:, 0, :] > 0]
    assert torch.allclose(kept, torch.full_like(kept, 1.0 / (1.0 - p)),
                          atol=1e-2)
    # drop fraction near p over B*F channels
    frac = (yf[:, 0, :] == 0).float().mean().item()
    assert abs(frac - p) < 0.08, frac
    # backward recomputes the identical mask: grad of sum(y) w.r.t. x is
    # the mask * scale
    from fmda_amd.ops.interface import _FusedSpatialDropout
    x2 = torch.ones(B, T, F, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y2 = _FusedSpatialDropout.apply(x2, p, 1234)
    y2.sum().backward()
    assert (y2.float() == x2.grad.float()).all()  # x==1 -> y == mask*scale
    # different seeds give different masks
    y3 = ext.spatial_dropout_fused(x, p, 99)
    assert not (y3.float() == yf).all()


def test_spatial_dropout_model_train_step():
    """The reference API default (spatial_dropout=True) runs the engine
    path end-to-end on GPU: forward + backward + fused optimizer."""
    from fmda_amd.models import BiGRU
    from fmda_amd.optim import FusedClipAdam
    torch.manual_seed(21)
    m = BiGRU(128, 96, 4, n_layers=2, spatial_dropout=True,
              dropout=0.3).cuda()
    opt = FusedClipAdam(m.parameters(), lr=1e-3, clip=50.0)
    x = (torch.randn(16, 30, 96, device="cuda") * 0.5).bfloat16()
    y = (torch.rand(16, 4, device="cuda") < 0.3).float()
    m.train()
    for _ in range(2):
        opt.zero_grad(set_to_none=True)
        logits = m(x)
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            logits.float(), y)
        loss.backward()
        opt.step()
    torch.cuda.synchronize()
    for prm in m.parameters():
        assert torch.isfinite(prm).all()


def test_gru_backward_h0_bf16_v3_kernel():
    """h0 BPTT on the bf16 Hp=128 v3 kernel specifically (swizzled h0
    staging + the dgh0 epilogue store feeding the dW_hh h0 term) — the
    fp32 h0 backward test exercises the v2 kernel only."""
    from fmda_amd.ops.interface import gru_directions
    torch.manual_seed(17)
    H, T, B, n_dir = 128, 20, 48, 2
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    h0 = (torch.randn(n_dir, B, H) * 0.5).cuda()
    dO = torch.randn(B, T, n_dir * H).cuda()
    dH = torch.randn(n_dir, B, H).cuda()

    gi1 = gi.bfloat16().requires_grad_(True)
    w1 = w.bfloat16().requires_grad_(True)
    b1 = bhh.clone().requires_grad_(True)
    h01 = h0.clone().requires_grad_(True)
    out1, hl1 = gru_directions(gi1, w1, b1, h01)
    ((out1.float() * dO).sum() + (hl1 * dH).sum()).backward()

    gi2 = gi.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = bhh.clone().requires_grad_(True)
    h02 = h0.clone().requires_grad_(True)
    out2, hl2 = _gru_ref_from_gi_autograd(gi2, w2, b2, h02)
    ((out2 * dO).sum() + (hl2 * dH).sum()).backward()

    assert (out1.float() - out2).abs().max() < 0.05
    for nm, a, b in [("dw", w1.grad.float(), w2.grad),
                     ("dbhh", b1.grad, b2.grad),
                     ("dh0", h01.grad, h02.grad)]:
        rel = (a - b).norm() / b.norm().clamp(min=1e-6)
        assert rel < 5e-2, (nm, float(rel))
    # the t=0 dW_hh term actually matters: recompute without it
    out3, hl3 = _gru_ref_from_gi_autograd(
        gi.clone().requires_grad_(True), w3 := w.clone().requires_grad_(True),
        bhh.clone().requires_grad_(True), None)
    ((out3 * dO).sum() + (hl3 * dH).sum()).backward()
    assert (w1.grad.float() - w3.grad).norm() / w3.grad.norm() > 1e-3


def test_gru_randomized_shape_sweep():
    """Seeded randomized sweep over (H, T, B, n_dir, dtype, h0?) — odd
    batch/sequence sizes and padded hidden sizes that the fixed
    parametrizations miss. Forward vs the golden recurrence; backward via
    gru_directions for the fp32 draws."""
    from fmda_amd.ops.interface import gru_directions
    import random
    ext = _ext()
    rng = random.Random(20240914)
    for trial in range(12):
        T = rng.randint(2, 40)
        B = rng.choice([1, 2, 3, 7, 17, 33, 65, 130])
        n_dir = rng.choice([1, 2])
        bf16 = rng.random() < 0.5
        # bf16 needs Hp >= 32 (MFMA K=32; Hp=16 is rejected at the binding)
        H = rng.choice([32, 64, 128, 256] if bf16 else [16, 32, 64, 128, 256])
        use_h0 = rng.random() < 0.5
        torch.manual_seed(1000 + trial)
        gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
        w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
        bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
        h0 = ((torch.randn(n_dir, B, H) * 0.5).cuda().contiguous()
              if use_h0 else None)
        if bf16:
            out, hlast = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh, h0)
            out_ref, hl_ref = _gru_ref_from_gi(gi, w, bhh, h0)
            ctx = (trial, H, T, B, n_dir, "bf16", use_h0)
            assert (out.float() - out_ref).abs().max() < 0.06, ctx
            assert (hlast - hl_ref).abs().max() < 0.06, ctx
        else:
            gi1 = gi.clone().requires_grad_(True)
            w1 = w.clone().requires_grad_(True)
            b1 = bhh.clone().requires_grad_(True)
            out, hlast = gru_directions(gi1, w1, b1, h0)
            dO = torch.randn_like(out)
            (out * dO).sum().backward()
            gi2 = gi.clone().requires_grad_(True)
            w2 = w.clone().requires_grad_(True)
            b2 = bhh.clone().requires_grad_(True)
            out2, _ = _gru_ref_from_gi_autograd(gi2, w2, b2, h0)
            (out2 * dO).sum().backward()
            ctx = (trial, H, T, B, n_dir, "fp32", use_h0)
            assert torch.allclose(out, out2, atol=2e-4), ctx
            for a, b in [(gi1.grad, gi2.grad), (w1.grad, w2.grad),
                         (b1.grad, b2.grad)]:
                scale = b.abs().max().clamp(min=1.0)
                assert ((a - b).abs().max() / scale) < 2e-4, ctx


@pytest.mark.parametrize("dtype", ["fp32", "bf16"])
@pytest.mark.parametrize("n_dir", [1, 2])
def test_gru_single_timestep_boundary(dtype, n_dir):
    """T=1: every boundary special-case (prologue==epilogue, dGh slot
    always out of range, h0 staging at the only step) collapses into one
    iteration — forward and full backward vs the golden recurrence."""
    from fmda_amd.ops.interface import gru_directions
    torch.manual_seed(31)
    H, B, T = 128, 33, 1
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    h0 = (torch.randn(n_dir, B, H) * 0.5).cuda().contiguous()
    cast = ((lambda t: t.bfloat16()) if dtype == "bf16"
            else (lambda t: t.clone()))

    gi1 = cast(gi).requires_grad_(True)
    w1 = cast(w).requires_grad_(True)
    b1 = bhh.clone().requires_grad_(True)
    h01 = h0.clone().requires_grad_(True)
    out1, hl1 = gru_directions(gi1, w1, b1, h01)
    dO = torch.randn(B, T, n_dir * H).cuda()
    (out1.float() * dO).sum().backward()

    gi2 = gi.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = bhh.clone().requires_grad_(True)
    h02 = h0.clone().requires_grad_(True)
    out2, hl2 = _gru_ref_from_gi_autograd(gi2, w2, b2, h02)
    (out2 * dO).sum().backward()

    tol = 5e-2 if dtype == "bf16" else 2e-4
    assert (out1.float() - out2).abs().max() < tol
    for nm, a, b in [("dgi", gi1.grad.float(), gi2.grad),
                     ("dw", w1.grad.float(), w2.grad),
                     ("dbhh", b1.grad, b2.grad),
                     ("dh0", h01.grad, h02.grad)]:
        scale = b.abs().max().clamp(min=1e-2)
        assert ((a - b).abs().max() / scale) < tol, (nm, dtype, n_dir)


def test_odd_hidden_size_model_matches_cpu():
    """Odd H (25): the paired pooling kernels cannot apply, so the model
    must take the eager pooling path on GPU and still match the CPU
    oracle end to end (fwd + grads)."""
    from fmda_amd.models import BiGRU
    torch.manual_seed(44)
    m = BiGRU(25, 20, 4, n_layers=1, spatial_dropout=False, dropout=0.0)
    x = torch.randn(9, 11, 20)
    logits_cpu = m(x)
    logits_cpu.sum().backward()
    gref = {n: p.grad.clone() for n, p in m.named_parameters()}

    mg = BiGRU(25, 20, 4, n_layers=1, spatial_dropout=False, dropout=0.0)
    mg.load_state_dict(m.state_dict())
    mg = mg.cuda()
    logits_gpu = mg(x.cuda())
    logits_gpu.sum().backward()
    assert (logits_gpu.cpu() - logits_cpu).abs().max() < 1e-3
    for n, p in mg.named_parameters():
        rel = (p.grad.cpu() - gref[n]).abs().max() / \
            gref[n].abs().max().clamp(min=1e-3)
        assert rel < 1e-3, n


def test_gru_unidirectional_h512_column_split():
    """n_dir=1 on the Hp=512 column-split kernels (their grid/direction
    mapping differs from the bidirectional case): fwd vs the golden
    recurrence and full backward vs fp32 autograd at rel-L2 tolerance."""
    from fmda_amd.ops.interface import gru_directions
    torch.manual_seed(51)
    H, T, B, n_dir = 512, 6, 33, 1
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    dO = torch.randn(B, T, n_dir * H).cuda()

    gi1 = gi.bfloat16().requires_grad_(True)
    w1 = w.bfloat16().requires_grad_(True)
    b1 = bhh.clone().requires_grad_(True)
    out1, _ = gru_directions(gi1, w1, b1)
    (out1.float() * dO).sum().backward()

    gi2 = gi.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = bhh.clone().requires_grad_(True)
    out2, _ = _gru_ref_from_gi_autograd(gi2, w2, b2)
    (out2 * dO).sum().backward()

    assert ((out1.float() - out2).norm() / out2.norm()) < 0.03
    for nm, a, b in [("dgi", gi1.grad.float(), gi2.grad),
                     ("dw", w1.grad.float(), w2.grad),
                     ("dbhh", b1.grad, b2.grad)]:
        rel = (a - b).norm() / b.norm().clamp(min=1e-3)
        assert rel < 8e-2, (nm, float(rel))


def test_fp32_h512_backward_raises_cleanly():
    from fmda_amd.ops.interface import gru_directions
    gi = torch.randn(4, 3, 3 * 512).cuda().requires_grad_(True)
    w = (torch.randn(1, 3 * 512, 512) * 0.1).cuda().requires_grad_(True)
    bhh = torch.randn(1, 3 * 512).cuda()
    out, _ = gru_directions(gi, w, bhh)
    import pytest as _pt
    with _pt.raises(RuntimeError, match="fp32 backward unsupported"):
        out.sum().backward()
