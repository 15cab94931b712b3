"""Per-shape microbench: sub-pixel deconv fwd / stride-2 bwd-data vs
MIOpen, on the exact FlowNetS batch-64 384x512 shapes.

    gpurun -- 'python tools/bench_subpixel.py | tee gpurun_out/subpixel.txt'
"""

import sys
import time

sys.path.insert(0, ".")

import torch  # noqa: E402

from deepof_amd.ops.deconv import (conv2d_bwd_data_subpixel,  # noqa: E402
                                   deconv2d_fwd)

DEV = "cuda:0"


def timeit(fn, n=20, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1000


def cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def bench_bwd_data(name, B, C, K, IH, IW, R, pad):
    OH, OW = (IH + 2 * pad - R) // 2 + 1, (IW + 2 * pad - R) // 2 + 1
    gy = cl(torch.randn(B, K, OH, OW, device=DEV, dtype=torch.bfloat16))
    w = cl(torch.randn(K, C, R, R, device=DEV, dtype=torch.bfloat16) * .05)
    x_shape = (B, C, IH, IW)

    def ours():
        return conv2d_bwd_data_subpixel(gy, w, pad, IH, IW)

    def mio():
        return torch.ops.aten.convolution_backward(
            gy, torch.empty(x_shape, device=DEV, dtype=torch.bfloat16)
            .contiguous(memory_format=torch.channels_last), w, None,
            [2, 2], [pad, pad], [1, 1], False, [0, 0], 1,
            [True, False, False])[0]

    t_h, t_m = timeit(ours), timeit(mio)
    err = (ours().float() - mio().float()).abs().max().item()
    scale = mio().float().abs().max().item()
    print(f"bwd2 {name:10s} C={C:4d} K={K:4d} {IH:3d}x{IW:3d} R={R} | "
          f"ours {t_h:7.3f} ms  miopen {t_m:7.3f} ms  x{t_m/t_h:5.2f} "
          f"relerr {err/max(scale,1e-6):.2e}")
    return t_h, t_m


def bench_deconv(name, B, C, K, IH, IW):
    x = cl(torch.randn(B, C, IH, IW, device=DEV, dtype=torch.bfloat16))
    w = torch.randn(C, K, 4, 4, device=DEV, dtype=torch.bfloat16) * .05
    b = torch.randn(K, device=DEV, dtype=torch.bfloat16)
    m = torch.nn.ConvTranspose2d(C, K, 4, 2, 1).to(DEV, torch.bfloat16)
    m = m.to(memory_format=torch.channels_last)
    with torch.no_grad():
        m.weight.copy_(w)
        m.bias.copy_(b)

    def ours():
        return deconv2d_fwd(x, w, b, act=1)

    def mio():
        return torch.nn.functional.elu(m(x))

    with torch.no_grad():
        t_h, t_m = timeit(ours), timeit(mio)
        err = (ours().float() - mio().float()).abs().max().item()
        scale = mio().float().abs().max().item()
    print(f"deconv {name:8s} C={C:4d} K={K:4d} {IH:3d}x{IW:3d}     | "
          f"ours {t_h:7.3f} ms  miopen {t_m:7.3f} ms  x{t_m/t_h:5.2f} "
          f"relerr {err/max(scale,1e-6):.2e}")
    return t_h, t_m


def main():
    B = 64
    torch.manual_seed(0)
    print("== stride-2 backward-data (FlowNetS @ 384x512 batch 64) ==")
    tot_h = tot_m = 0.0
    for args in [("conv2", B, 64, 128, 192, 256, 5, 2),
                 ("conv3_1", B, 128, 256, 96, 128, 5, 2),
                 ("conv4_1", B, 256, 512, 48, 64, 3, 1),
                 ("conv5_1", B, 512, 512, 24, 32, 3, 1),
                 ("conv6_1", B, 512, 1024, 12, 16, 3, 1)]:
        h, m_ = bench_bwd_data(*args)
        tot_h += h
        tot_m += m_
    print(f"   total: ours {tot_h:.3f} ms vs miopen {tot_m:.3f} ms")

    print("== deconv forward (decoder upconvs, padded concat) ==")
    tot_h = tot_m = 0.0
    for args in [("up1", B, 1024, 512, 6, 8),
                 ("up2", B, 1088, 256, 12, 16),
                 ("up3", B, 832, 128, 24, 32),
                 ("up4", B, 448, 64, 48, 64),
                 ("up5", B, 256, 32, 96, 128)]:
        h, m_ = bench_deconv(*args)
        tot_h += h
        tot_m += m_
    print(f"   total: ours {tot_h:.3f} ms vs miopen {tot_m:.3f} ms")


if __name__ == "__main__":
    main()
