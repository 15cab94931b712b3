"""Per-shape correctness + microbenchmark: MFMA conv2d_fwd vs MIOpen.

Run on a GPU box:
    python tools/bench_conv.py
"""

import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from deepof_amd.ops.functional import require_hip

# FlowNetS @ 384x512 batch 64 encoder/decoder conv shapes:
# (name, B, C, H, W, K, R, stride)
SHAPES = [
    ("conv2",   64,  64, 192, 256, 128, 5, 2),
    ("conv3_1", 64, 128,  96, 128, 256, 5, 2),
    ("conv3_2", 64, 256,  48,  64, 256, 3, 1),
    ("conv4_1", 64, 256,  48,  64, 512, 3, 2),
    ("conv4_2", 64, 512,  24,  32, 512, 3, 1),
    ("conv5_1", 64, 512,  24,  32, 512, 3, 2),
    ("conv5_2", 64, 512,  12,  16, 512, 3, 1),
    ("conv6_1", 64, 512,  12,  16, 1024, 3, 2),
    ("conv6_2", 64, 1024,  6,   8, 1024, 3, 1),
    ("pr2",     64, 194,  96, 128, 2, 3, 1),
    ("pr1",     64,  98, 192, 256, 2, 3, 1),
]


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000.0


def main():
    hip = require_hip()
    torch.backends.cudnn.benchmark = True
    dev = "cuda:0"
    print(f"{'layer':9s} {'o128 ms':>9s} {'o256 ms':>9s} {'miopen ms':>10s} "
          f"{'speedup':>8s} {'TF/s':>7s}  maxerr")
    for name, B, C, H, W, K, R, stride in SHAPES:
        pad = R // 2
        torch.manual_seed(0)
        x = (torch.randn(B, C, H, W, device=dev, dtype=torch.bfloat16)
             .to(memory_format=torch.channels_last))
        w = (torch.randn(K, C, R, R, device=dev, dtype=torch.bfloat16)
             .to(memory_format=torch.channels_last)) * (1.0 / (C * R * R) ** 0.5)
        b = torch.randn(K, device=dev, dtype=torch.float32) * 0.1

        def miopen():
            y = F.conv2d(x, w, b.bfloat16(), stride=stride, padding=pad)
            return F.elu(y)

        def ours():
            return hip.conv2d_fwd(x, w, b, stride, pad, 1)

        def ours256():
            return hip.conv2d_fwd256(x, w, b, stride, pad, 1)

        use256 = C % 32 == 0 and K >= 192

        try:
            y1 = ours()
        except Exception as e:
            print(f"{name:9s} OURS FAILED: {e}")
            continue
        y0 = miopen()
        # fp32 reference for error norm
        ref = F.elu(F.conv2d(x.float(), w.float(), b, stride=stride,
                             padding=pad))
        err_ours = (y1.float() - ref).abs().max().item()
        err_mio = (y0.float() - ref).abs().max().item()
        err_256 = float('nan')
        t_256 = float('nan')
        if use256:
            y2 = ours256()
            err_256 = (y2.float() - ref).abs().max().item()
            t_256 = bench(ours256)
        t_ours = bench(ours)
        t_mio = bench(miopen)
        oh = (H + 2 * pad - R) // stride + 1
        ow = (W + 2 * pad - R) // stride + 1
        flops = 2.0 * B * oh * ow * K * C * R * R
        tbest = min(t_ours, t_256) if use256 else t_ours
        tf = flops / (tbest / 1000) / 1e12
        print(f"{name:9s} {t_ours:9.3f} {t_256:9.3f} {t_mio:10.3f} "
              f"{t_mio/tbest:8.2f} {tf:7.1f}  ours={err_ours:.3e} "
              f"o256={err_256:.3e} miopen={err_mio:.3e}")


if __name__ == "__main__":
    main()
