"""Fused conv+bias+activation module with measured per-shape dispatch.

On GPU, the hand-written MFMA implicit-GEMM kernel (conv_mfma.hip) and
the MIOpen path are BOTH timed on the first occurrence of each conv
shape; the faster one is cached and used from then on ("measure, don't
guess" — the MI355X dispatch rule).  The HIP kernel additionally fuses
bias + ELU/LeakyReLU into the GEMM epilogue, which also removes
PyTorch's separate elementwise activation kernels from the hot path.

Backward: activation gradient is folded analytically from the OUTPUT
(ELU'(pre) = 1 if y > 0 else y + 1), then aten.convolution_backward
produces input/weight/bias grads.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

_ACT_CODE = {None: 0, "none": 0, "elu": 1, "leaky_relu": 2, "relu": 3}
_CODE_ACT = {0: "none", 1: "elu", 2: "leaky_relu", 3: "relu"}
_dispatch_cache: dict[tuple, str] = {}
_wrw_cache: dict[tuple, str] = {}


def _act(y, act: str | None):
    if act == "elu":
        return F.elu(y)
    if act == "leaky_relu":
        return F.leaky_relu(y, 0.1)
    if act == "relu":
        return F.relu(y)
    return y


class _FusedConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, act_code, variant="hip128"):
        from .functional import require_hip

        hip = require_hip()
        fn = hip.conv2d_fwd256 if variant == "hip256" else hip.conv2d_fwd
        y = fn(x, w, bias if bias is not None else torch.Tensor(),
               stride, pad, act_code)
        ctx.save_for_backward(x, w, y)
        ctx.meta = (stride, pad, act_code, bias is not None)
        return y

    @staticmethod
    def backward(ctx, gy):
        return _conv_backward_impl(ctx, gy) + (None,)


class _MiopenConvFn(torch.autograd.Function):
    """MIOpen forward (it measured faster for this shape), but OUR
    backward dispatch: fused act-grad + MFMA backward-data kernels."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, pad, act_code):
        y = torch.ops.aten.convolution(
            x, w, bias, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1)
        y = _act(y, _CODE_ACT[act_code])
        y = y.contiguous(memory_format=torch.channels_last)
        ctx.save_for_backward(x, w, y)
        ctx.meta = (stride, pad, act_code, bias is not None)
        return y

    @staticmethod
    def backward(ctx, gy):
        return _conv_backward_impl(ctx, gy)


def _conv_backward_impl(ctx, gy):
    x, w, y = ctx.saved_tensors
    stride, pad, act_code, has_bias = ctx.meta
    gy = gy.contiguous(memory_format=torch.channels_last)
    if act_code in (1, 2, 3):
        from .functional import require_hip

        # one fused pass (gy * act_grad(y)) instead of where+mul
        gy = require_hip().act_grad(gy, y, act_code)

    need_gx = ctx.needs_input_grad[0]
    gx = None
    k = w.shape[-1]
    # stride-1 backward-data IS a forward conv with the transposed,
    # spatially flipped weight (dx = gy * W^T_rot) -> reuse the MFMA
    # kernel when eligible (K*k*k % 64 == 0, same-size output)
    if (need_gx and stride == 1 and w.shape[0] % 8 == 0
            and (w.shape[0] * k * k) % 64 == 0
            and x.shape[-2:] == gy.shape[-2:]):
        from .functional import require_hip

        wt = (w.transpose(0, 1).flip(-1, -2)
              .contiguous(memory_format=torch.channels_last))
        gx = require_hip().conv2d_fwd(gy, wt, torch.Tensor(), 1, pad, 0)
        need_gx = False
    # stride-2 backward-data: zero-insertion-free sub-pixel
    # decomposition (4 parity launches of the strided-out MFMA
    # kernel), adopted per shape when it measures faster than MIOpen
    if need_gx and stride == 2:
        from .deconv import bwd_data_dispatch

        gx = bwd_data_dispatch(gy, x.shape, w, stride, pad)
        if gx is not None:
            need_gx = False

    # weight grad via the MFMA wrw2 kernel (natural-layout staging
    # + ds_read_b64_tr_b16 transpose reads) when it measures faster
    # than MIOpen for this shape (tools/bench_wrw2.py; it wins on
    # the small-spatial conv6-class shapes).  DEEPOF_WRW=0 disables.
    import os as _os

    need_gw = ctx.needs_input_grad[1]
    gw = None
    if (need_gw and _os.environ.get("DEEPOF_WRW") != "0"
            and x.shape[1] % 64 == 0 and w.shape[0] % 8 == 0):
        gw = _maybe_hip_wrw(gy, x, w, stride, pad)
        if gw is not None:
            need_gw = False

    gx2, gw2, gb = torch.ops.aten.convolution_backward(
        gy, x, w, [w.shape[0]] if has_bias else None,
        [stride, stride], [pad, pad], [1, 1], False, [0, 0], 1,
        [need_gx, need_gw, has_bias and ctx.needs_input_grad[2]],
    )
    if gx is None:
        gx = gx2
    if gw is None:
        gw = gw2
    return gx, gw, gb, None, None, None


class FusedConvAct(nn.Module):
    """Conv2d + bias + activation; HIP MFMA kernel on GPU when it wins."""

    def __init__(self, cin, cout, k=3, stride=1, act: str | None = "elu"):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride=stride, padding=k // 2,
                              bias=True)
        self.act_name = act
        self.act_code = _ACT_CODE[act]
        self.cin, self.k, self.stride = cin, k, stride

    def _hip_eligible(self, x):
        return (
            x.is_cuda
            and self.cin % 8 == 0
            and (self.k * self.k * self.cin) % 64 == 0
            and x.is_contiguous(memory_format=torch.channels_last)
        )

    def _miopen(self, x):
        return _act(self.conv(x), self.act_name)

    def _hip(self, x, variant="hip128"):
        w = self.conv.weight
        if torch.is_autocast_enabled():
            x = x.to(torch.bfloat16)
            w = w.to(torch.bfloat16)
        if x.dtype != torch.bfloat16 or w.dtype != torch.bfloat16:
            return None
        w = w.contiguous(memory_format=torch.channels_last)
        x = x.contiguous(memory_format=torch.channels_last)
        return _FusedConvFn.apply(x, w, self.conv.bias, self.stride,
                                  self.k // 2, self.act_code, variant)

    def _miopen_fn(self, x):
        """MIOpen fwd with OUR backward (fused act-grad + sub-pixel
        stride-2 backward-data dispatch)."""
        w = self.conv.weight
        if torch.is_autocast_enabled():
            x = x.to(torch.bfloat16)
            w = w.to(torch.bfloat16)
        if x.dtype != torch.bfloat16 or w.dtype != torch.bfloat16:
            return None
        w = w.contiguous(memory_format=torch.channels_last)
        x = x.contiguous(memory_format=torch.channels_last)
        return _MiopenConvFn.apply(x, w, self.conv.bias, self.stride,
                                   self.k // 2, self.act_code)

    def forward(self, x):
        if not self._hip_eligible(x):
            return self._miopen(x)
        key = (self.cin, self.conv.out_channels, self.k, self.stride,
               tuple(x.shape), self.act_code)
        choice = _dispatch_cache.get(key)
        if choice is None:
            choice = self._autotune(x, key)
        if choice.startswith("hip"):
            y = self._hip(x, choice)
            if y is not None:
                return y
        # MIOpen won the forward; still claim the backward for stride-2
        # layers where the sub-pixel bwd-data kernel is eligible
        import os as _os

        if (self.stride == 2 and self.conv.out_channels % 64 == 0
                and torch.is_grad_enabled()
                and not _os.environ.get("DEEPOF_NO_MIOPENFN")):
            y = self._miopen_fn(x)
            if y is not None:
                return y
        return self._miopen(x)

    @torch.no_grad()
    def _autotune(self, x, key) -> str:
        import time

        y = self._hip(x)
        if y is None:
            _dispatch_cache[key] = "miopen"
            return "miopen"

        def timeit(fn, n=6):
            fn()  # warm
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            return time.perf_counter() - t0

        cands = {"hip128": lambda: self._hip(x, "hip128"),
                 "miopen": lambda: self._miopen(x)}
        if self.cin % 32 == 0 and self.conv.out_channels >= 192:
            cands["hip256"] = lambda: self._hip(x, "hip256")
        times = {}
        for name, fn in cands.items():
            try:
                times[name] = timeit(fn)
            except Exception:
                pass
        choice = min(times, key=times.get)
        _dispatch_cache[key] = choice
        return choice


def _maybe_hip_wrw(gy, x, w, stride, pad):
    """Measured dispatch for the weight gradient: time the MFMA wrw
    kernel against aten.convolution_backward's wrw once per shape."""
    import time

    from .functional import require_hip

    key = ("wrw", tuple(x.shape), tuple(w.shape), stride)
    choice = _wrw_cache.get(key)
    hip = require_hip()
    R, S = w.shape[2], w.shape[3]
    if choice is None:
        def ours():
            return hip.conv2d_wrw2(gy, x, R, S, stride, pad)

        def mio():
            return torch.ops.aten.convolution_backward(
                gy, x, w, None, [stride, stride], [pad, pad], [1, 1],
                False, [0, 0], 1, [False, True, False])[1]

        def timeit(fn, n=4):
            fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            return time.perf_counter() - t0

        with torch.no_grad():
            t_h = timeit(ours)
            t_m = timeit(mio)
        choice = "hip" if t_h < t_m else "miopen"
        _wrw_cache[key] = choice
    if choice == "hip":
        return hip.conv2d_wrw2(gy, x, R, S, stride, pad)
    return None
