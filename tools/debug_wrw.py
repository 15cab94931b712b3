import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepof_amd.ops.functional import require_hip
hip = require_hip()
dev = "cuda:0"

def check(name, B, C, H, W, K, R, stride, seed=0):
    pad = R // 2
    OH, OW = (H+2*pad-R)//stride+1, (W+2*pad-R)//stride+1
    torch.manual_seed(seed)
    x = (torch.randn(B, C, H, W, device=dev, dtype=torch.bfloat16)
         .to(memory_format=torch.channels_last))
    gy = (torch.randn(B, K, OH, OW, device=dev, dtype=torch.bfloat16)
          .to(memory_format=torch.channels_last))
    w = (torch.empty(K, C, R, R, device=dev, dtype=torch.bfloat16)
         .to(memory_format=torch.channels_last))
    got = hip.conv2d_wrw(gy, x, R, R, stride, pad).float()
    want = torch.ops.aten.convolution_backward(
        gy, x, w, None, [stride]*2, [pad]*2, [1,1], False, [0,0], 1,
        [False, True, False])[1].float()
    err = (got - want).abs()
    rel = err.max().item() / (want.abs().max().item() + 1e-9)
    print(f"{name}: rel={rel:.3e}", "OK" if rel < 2e-2 else "BAD")
    if rel >= 2e-2:
        # locate: which output elements wrong
        bad = (err > 0.05 * want.abs().max()).nonzero()
        print("  bad count", bad.shape[0], "of", want.numel(),
              "first:", bad[:5].tolist())
        print("  got ", got.flatten()[:6].tolist())
        print("  want", want.flatten()[:6].tolist())

check("1x1 single-block", 1, 64, 8, 8, 64, 1, 1)      # M=64, 1 tile
check("1x1 two-stage", 1, 64, 16, 8, 64, 1, 1)        # M=128, 2 stages
check("3x3", 2, 64, 16, 16, 64, 3, 1)
check("stride2 5x5", 2, 64, 32, 32, 128, 5, 2)
