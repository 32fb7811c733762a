"""Localize the conv_dw_slab numerics bug: per-(kh,kw) / per-ci diff map."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from distributedmnist_amd import _C
from distributedmnist_amd.ops import cpu_ref

ext = _C.ext()
torch.manual_seed(4)
bf16 = torch.bfloat16
NB, H, W, Cin, Cout = int(os.environ.get('DBG_NB', 2)), 14, 14, 32, 64
x = (torch.rand(NB, H, W, Cin) - 0.5).to(bf16).float()
w = (torch.randn(5, 5, Cin, Cout) * 0.1).to(bf16).float()
b = torch.randn(Cout) * 0.1
y, amax = cpu_ref.conv_pool_fwd(x, w, b)
dy = (torch.randn(NB, H // 2, W // 2, Cout) * 0.1).to(bf16).float()
dx_ref, dw_ref, db_ref = cpu_ref.conv_pool_bwd(dy, x, w, y, amax)

g = lambda t: t.to(device="cuda", dtype=bf16).contiguous()
yg, amaxg = ext.conv_pool_fwd(g(x), g(w), b.cuda().float())
agree = (amaxg.cpu() == amax).float().mean().item()
print(f"amax agreement: {agree:.6f}")
# recompute the CPU reference FROM THE GPU's own argmax/y so tie-routing
# differences (both valid subgradients) don't pollute the comparison
dx_ref, dw_ref, db_ref = cpu_ref.conv_pool_bwd(
    dy, x, w, yg.cpu().float(), amaxg.cpu())
dx, dw, db = ext.conv_pool_bwd(g(dy), g(x), g(w), yg, amaxg, True)
dwc = dw.cpu()
d = (dwc - dw_ref).abs()
print("dw max diff:", float(d.max()), " ref scale:", float(dw_ref.abs().max()))
print("per (kh,kw) max diff:")
for kh in range(5):
    print("  ", [f"{float(d[kh, kw].max()):.4f}" for kw in range(5)])
print("per-ci max (kh=2,kw=2):", [f"{float(d[2,2,ci].max()):.3f}" for ci in range(0, 32, 4)])
print("sample ref vs got at [0,0,0,:4]:", dw_ref[0,0,0,:4].tolist(), dwc[0,0,0,:4].tolist())
print("sample ref vs got at [2,2,0,:4]:", dw_ref[2,2,0,:4].tolist(), dwc[2,2,0,:4].tolist())
print("dx max diff:", float((dx.cpu().float()-dx_ref).abs().max()))
