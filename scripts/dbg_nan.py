import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from fmda_amd.ops import load_extension
ext = load_extension()
torch.manual_seed(5)
H, F, B, T, D = 512, 96, 48, 2, 2
gi = (torch.randn(B, T, D*3*H, device="cuda") * 0.5).bfloat16()
w = (torch.randn(D, 3*H, H, device="cuda") * 0.2).bfloat16()
bhh = (torch.randn(D, 3*H, device="cuda") * 0.1).float()
out, hl = ext.gru_fwd(gi, w, bhh)
z_do = torch.zeros_like(out); z_h = torch.zeros_like(hl)
for it in range(3):
    dgi, dgh, dh0, dbhh, dbih = ext.gru_bwd(gi, w, bhh, out, z_do, z_h)
    m = dgi.isnan()
    nz = m.nonzero()
    print(f"run{it}: nan={int(m.sum())} nonzero_abs={int((dgi.float().abs()>0).sum())}")
    if it == 0:
        print("first coords (b,t,col):", nz[:6].tolist())
        b0, t0, c0 = nz[0].tolist()
        chunk = dgi[b0, t0, c0//8*8:(c0//8*8+8)].float().tolist()
        print("16B chunk around first:", chunk)
        # lane structure: col%32 histogram of nan
        print("col%32 hist:", torch.bincount(nz[:,2] % 32, minlength=32).tolist())
        print("b%16 hist:", torch.bincount(nz[:,0] % 16, minlength=16).tolist())
