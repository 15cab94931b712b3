"""wrw2 numerics (per-tap max error vs fp32 autograd) + perf vs MIOpen.

    gpurun -- 'python tools/bench_wrw2.py'
"""

import sys
import time

sys.path.insert(0, ".")

import torch

DEV = "cuda:0"


def cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def timeit(fn, n=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000


def check(name, B, C, K, IH, IW, R, stride, pad, perf=True):
    from deepof_amd.ops.functional import require_hip

    hip = require_hip()
    torch.manual_seed(0)
    x = torch.randn(B, C, IH, IW, device=DEV)
    w = torch.randn(K, C, R, R, device=DEV).requires_grad_(True)
    y = torch.nn.functional.conv2d(x, w, stride=stride, padding=pad)
    gy = torch.randn_like(y)
    (gw_ref,) = torch.autograd.grad(y, w, gy)

    gyb, xb = cl(gy.bfloat16()), cl(x.bfloat16())

    def ours():
        return hip.conv2d_wrw2(gyb, xb, R, R, stride, pad)

    def mio():
        return torch.ops.aten.convolution_backward(
            gyb, xb, cl(w.detach().bfloat16()), None, [stride, stride],
            [pad, pad], [1, 1], False, [0, 0], 1, [False, True, False])[1]

    got = ours().float()
    scale = gw_ref.abs().max().item()
    err = (got - gw_ref).abs()
    print(f"{name}: rel err overall {err.max().item()/scale:.2e}")
    # per-tap max error
    per_tap = err.amax(dim=(0, 1))
    bad = (per_tap / scale > 0.05).nonzero()
    if len(bad):
        print("  taps with rel err > 5%:", bad.tolist())
        for r, s in bad.tolist()[:4]:
            print(f"   tap({r},{s}): max err {per_tap[r, s].item():.4f}")
    if perf:
        t2, tm = timeit(ours), timeit(mio)
        print(f"  v2 {t2:7.3f} ms  miopen {tm:7.3f} ms"
              f"  v2 speedup x{tm/t2:.2f}")


def main():
    # small correctness shapes (incl. the v1 failing case)
    check("small s1", 2, 64, 16, 16, 20, 3, 1, 1, perf=False)
    check("small s2", 2, 64, 32, 20, 28, 5, 2, 2, perf=False)
    B = 64
    # FlowNetS batch-64 @384x512 wrw shapes
    for args in [("conv2", B, 64, 128, 192, 256, 5, 2, 2),
                 ("conv3_1", B, 128, 256, 96, 128, 5, 2, 2),
                 ("conv3_2", B, 256, 256, 48, 64, 3, 1, 1),
                 ("conv4_1", B, 256, 512, 48, 64, 3, 2, 1),
                 ("conv4_2", B, 512, 512, 24, 32, 3, 1, 1),
                 ("conv5_1", B, 512, 512, 24, 32, 3, 2, 1),
                 ("conv6_1", B, 512, 1024, 12, 16, 3, 2, 1),
                 ("conv6_2", B, 1024, 1024, 6, 8, 3, 1, 1)]:
        check(*args)


if __name__ == "__main__":
    main()
