"""Correctness + timing: MFMA wrw kernel vs aten.convolution_backward."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepof_amd.ops.functional import require_hip

SHAPES = [
    ("conv2",   64,  64, 192, 256, 128, 5, 2),
    ("conv3_1", 64, 128,  96, 128, 256, 5, 2),
    ("conv3_2", 64, 256,  48,  64, 256, 3, 1),
    ("conv4_2", 64, 512,  24,  32, 512, 3, 1),
    ("conv5_2", 64, 512,  12,  16, 512, 3, 1),
    ("conv6_2", 64, 1024,  6,   8, 1024, 3, 1),
]

def bench(fn, it=10, wu=3):
    for _ in range(wu): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(it): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it*1000

hip = require_hip()
dev = "cuda:0"
print(f"{'layer':9s} {'ours ms':>8s} {'mio ms':>8s} {'spd':>5s} {'TF/s':>7s} maxrel")
for name, B, C, H, W, K, R, stride in SHAPES:
    pad = R // 2
    OH, OW = (H+2*pad-R)//stride+1, (W+2*pad-R)//stride+1
    torch.manual_seed(0)
    x = (torch.randn(B, C, H, W, device=dev, dtype=torch.bfloat16)
         .to(memory_format=torch.channels_last))
    gy = (torch.randn(B, K, OH, OW, device=dev, dtype=torch.bfloat16)
          .to(memory_format=torch.channels_last)) * 0.01
    w = (torch.empty(K, C, R, R, device=dev, dtype=torch.bfloat16)
         .to(memory_format=torch.channels_last))
    def ours(): return hip.conv2d_wrw(gy, x, R, R, stride, pad)
    def mio():
        return torch.ops.aten.convolution_backward(
            gy, x, w, None, [stride, stride], [pad, pad], [1, 1],
            False, [0, 0], 1, [False, True, False])[1]
    try:
        got = ours().float()
    except Exception as e:
        print(f"{name:9s} FAILED: {e}"); continue
    want = mio().float()
    scale = want.abs().max().item() + 1e-9
    err = (got - want).abs().max().item() / scale
    t_o, t_m = bench(ours), bench(mio)
    fl = 2.0 * B*OH*OW*K*C*R*R
    print(f"{name:9s} {t_o:8.3f} {t_m:8.3f} {t_m/t_o:5.2f} "
          f"{fl/(min(t_o,t_m)/1e3)/1e12:7.1f} {err:.3e}")
