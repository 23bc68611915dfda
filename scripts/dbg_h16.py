import itertools
import sys

import torch

sys.path.insert(0, ".")
from fmda_amd.ops import load_extension
from tests.test_gpu_kernels import _gru_ref_from_gi

ext = load_extension()
for H, n_dir, use_h0, B, T in itertools.product(
        [16, 32], [1, 2], [False, True], [32, 33], [8, 36]):
    torch.manual_seed(5)
    gi = (torch.randn(B, T, n_dir * 3 * H) * 0.5).cuda()
    w = (torch.randn(n_dir, 3 * H, H) * 0.2).cuda()
    bhh = (torch.randn(n_dir, 3 * H) * 0.1).cuda()
    h0 = (torch.randn(n_dir, B, H) * 0.5).cuda().contiguous() if use_h0 else None
    if H < 32:
        try:
            ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh, h0)
            print(f"FAIL H={H}: bf16 Hp<32 not rejected")
        except RuntimeError:
            print(f"ok   H={H} d={n_dir} h0={int(use_h0)} B={B} T={T} rejected")
        continue
    out, hl = ext.gru_fwd(gi.bfloat16(), w.bfloat16(), bhh, h0)
    out_ref, hl_ref = _gru_ref_from_gi(gi, w, bhh, h0)
    err = float((out.float() - out_ref).abs().max())
    flag = "FAIL" if err > 0.06 else "ok  "
    print(f"{flag} H={H} d={n_dir} h0={int(use_h0)} B={B} T={T} err={err:.4f}")
