import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepof_amd.ops.functional import require_hip
hip = require_hip()
dev = "cuda:0"

B, C, H, W, K, R, stride, pad = 1, 64, 8, 8, 64, 1, 1, 0
M = B * H * W

def wrw(gy, x):
    return hip.conv2d_wrw(gy, x, R, R, stride, pad).float()[:, :, 0, 0]

def ref(gy, x):
    gyf = gy.float().permute(0, 2, 3, 1).reshape(M, K)
    xf = x.float().permute(0, 2, 3, 1).reshape(M, C)
    return gyf.t() @ xf

def mk(t):
    return t.to(dev, torch.bfloat16).contiguous(
        memory_format=torch.channels_last)

torch.manual_seed(0)
xr = mk(torch.randn(B, C, H, W))
gr = mk(torch.randn(B, K, H, W))

# probe 1: x = ones -> D[k][c] = sum_p gy[p][k], const across c
x1 = mk(torch.ones(B, C, H, W))
d = wrw(gr, x1); want = ref(gr.cpu(), x1.cpu()).to(dev)
print("P1 x=1:", (d - want).abs().max().item(), "/", want.abs().max().item())

# probe 2: gy = ones -> D[k][c] = sum_p x[p][c], const across k
g1 = mk(torch.ones(B, K, H, W))
d = wrw(g1, xr); want = ref(g1.cpu(), xr.cpu()).to(dev)
print("P2 gy=1:", (d - want).abs().max().item(), "/", want.abs().max().item())

# probe 3: gy = delta at pixel 5, channel k arbitrary -> D[k][c]=gy[5][k]*x[5][c]
g3 = torch.zeros(B, K, H, W); g3[0, :, 0, 5] = torch.arange(K) % 7 - 3.0
g3 = mk(g3)
d = wrw(g3, xr); want = ref(g3.cpu(), xr.cpu()).to(dev)
print("P3 delta-pix:", (d - want).abs().max().item(), "/", want.abs().max().item())

# probe 4: single nonzero at gy[p=5][k=3], x[p=5][c=9]=1 else 0
g4 = torch.zeros(B, K, H, W); g4[0, 3, 0, 5] = 2.0
x4 = torch.zeros(B, C, H, W); x4[0, 9, 0, 5] = 1.0
d = wrw(mk(g4), mk(x4))
nz = d.nonzero()
print("P4 single: nonzero at", nz.tolist()[:5], "value",
      d[3, 9].item(), "(want D[3][9]=2)")

# probe 5: full random, dump 8x8 corner
d = wrw(gr, xr); want = ref(gr.cpu(), xr.cpu()).to(dev)
print("P5 rand err:", (d - want).abs().max().item())
print("got row0 :", [round(v,2) for v in d[0, :8].tolist()])
print("want row0:", [round(v,2) for v in want[0, :8].tolist()])
print("got col0 :", [round(v,2) for v in d[:8, 0].tolist()])
print("want col0:", [round(v,2) for v in want[:8, 0].tolist()])
