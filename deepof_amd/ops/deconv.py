"""Sub-pixel (zero-insertion-free) transposed convolution on the MFMA
conv kernel: stride-2 deconv forward and stride-2 conv backward-data.

A stride-2 transposed conv scatters each input pixel into a 2x2 output
neighborhood.  Instead of dilating with zeros (what im2col/MIOpen-style
transposed convs pay 4x MFMA work for), each OUTPUT PARITY (oy%2, ox%2)
is an independent stride-1 convolution with a small sub-filter of the
taps that actually land on that parity.  Four parity launches of
ops/hip/conv_mfma.hip's strided-output kernel write disjoint pixels of
the full-resolution output.

This one primitive is both
  - the decoder's 4x4/s2 deconv forward (fused bias+ELU) — reference
    /root/reference/flyingChairsWrapFlow.py:65-66 with the bilinear
    init of flyingChairsTrain.py:78-92, and
  - backward-data of every stride-2 encoder conv (the single largest
    MIOpen chunk of the r01 step: 7.9 of 30.5 ms,
    profiles/r01_flownets_bench_steady_state.md).
"""

from __future__ import annotations

import time

import torch

_EMPTY = None


def _empty():
    global _EMPTY
    if _EMPTY is None:
        _EMPTY = torch.Tensor()
    return _EMPTY


def _axis_plan(R: int, pad: int):
    """Per output parity q of one axis: (offset u, descending tap list,
    stride-1 pad) such that out(2t+u) = sum_j x(t + j - pad_p) * w[taps[j]].

    Derivation: out(iy) = sum_{r == (iy+pad) mod 2} in((iy+pad-r)/2)*w(r)
    with iy = 2t+u, u = (q-pad) mod 2; ascending input position j maps to
    descending tap r = r_max - 2j and pad_p = (r_max - u - pad) / 2.
    """
    plans = []
    for q in (0, 1):
        u = (q - pad) % 2
        taps = list(range(q, R, 2))
        if not taps:
            plans.append((u, [], 0))
            continue
        r_max = taps[-1]
        pad_p = (r_max - u - pad) // 2
        plans.append((u, taps[::-1], pad_p))
    return plans


def subpixel_eligible(red_ch: int, out_ch: int) -> bool:
    """red_ch: reduction channels (C_in of the transposed conv);
    out_ch: output channels.  Parity sub-filters have odd tap counts, so
    the kernel's (R*S*C) % 64 requirement needs red_ch % 64 == 0."""
    return red_ch % 64 == 0 and out_ch >= 8


_tab_cache: dict[tuple, tuple] = {}


def _plan_tabs(R: int, S: int, pad: int, N: int, M: int, device):
    """Device int32 tables for the single-launch path: pack rows
    {off, nry, nrx, ty0..3, tx0..3, 0} and conv rows
    {woff, R', S', pad_y, pad_x, off_y, off_x, 0} per parity."""
    key = (R, S, pad, N, M, str(device))
    cached = _tab_cache.get(key)
    if cached is not None:
        return cached
    pack_rows, conv_rows = [], []
    off = 0
    for uy, taps_y, pad_y in _axis_plan(R, pad):
        for ux, taps_x, pad_x in _axis_plan(S, pad):
            nry, nrx = len(taps_y), len(taps_x)
            assert 1 <= nry <= 4 and 1 <= nrx <= 4
            pack_rows.append([off, nry, nrx]
                             + taps_y + [0] * (4 - nry)
                             + taps_x + [0] * (4 - nrx) + [0])
            conv_rows.append([off, nry, nrx, pad_y, pad_x, uy, ux, 0])
            off += N * nry * nrx * M
    pack_tab = torch.tensor(pack_rows, dtype=torch.int32, device=device)
    conv_tab = torch.tensor(conv_rows, dtype=torch.int32, device=device)
    _tab_cache[key] = (pack_tab, conv_tab)
    return pack_tab, conv_tab


def conv_transpose2d_subpixel(x: torch.Tensor, w: torch.Tensor,
                              bias, pad: int, out_h: int, out_w: int,
                              act: int = 0,
                              out: torch.Tensor | None = None,
                              out_coff: int = 0) -> torch.Tensor:
    """y[b,n,oy,ox] = act(sum_{m,r,s} x[b,m,t,u] w[m,n,r,s] + bias[n])
    with oy = 2*t - pad + r (transposed-conv scatter semantics).

    x: [B, M, H, W] channels_last bf16; w: [M, N, R, S] bf16 — the
    conv2d weight itself for backward-data, the ConvTranspose2d weight
    for a deconv forward.  out: optional pre-allocated
    [B, >=out_coff+N, out_h, out_w] channels_last bf16 buffer (parities
    cover every pixel, no init needed for the full channel range).
    """
    from .functional import require_hip

    hip = require_hip()
    M, N, R, S = w.shape
    B = x.shape[0]
    if out is None:
        out = torch.empty(
            (B, N, out_h, out_w), device=x.device, dtype=torch.bfloat16
        ).contiguous(memory_format=torch.channels_last)
    b = bias if bias is not None else _empty().to(x.device)
    # every dispatcher pre-gates on subpixel_eligible (M % 64 == 0, the
    # kernel's reduction-tile requirement), so the single-launch path is
    # the only one: 1 pack launch + 1 all-parity conv launch
    assert M % 64 == 0, "gate with subpixel_eligible() first"
    wcl = w.contiguous(memory_format=torch.channels_last)
    pack_tab, conv_tab = _plan_tabs(R, S, pad, N, M, x.device)
    wp = hip.subpixel_pack(wcl, pack_tab, R, S)
    hip.conv2d_fwd_subpixel4(x, wp, b, out, conv_tab, N, act, 2, out_coff)
    return out


def deconv2d_fwd(x: torch.Tensor, w_ct: torch.Tensor, bias,
                 act: int = 0, out: torch.Tensor | None = None,
                 out_coff: int = 0) -> torch.Tensor:
    """Forward of nn.ConvTranspose2d(cin, cout, 4, stride=2, padding=1)
    with fused bias + activation.  w_ct: [C_in, C_out, 4, 4]."""
    H, W = x.shape[-2:]
    return conv_transpose2d_subpixel(
        x, w_ct, bias, 1, 2 * H, 2 * W, act, out=out, out_coff=out_coff)


def conv2d_bwd_data_subpixel(gy: torch.Tensor, w: torch.Tensor,
                             pad: int, ih: int, iw: int) -> torch.Tensor:
    """dx of a stride-2 conv: dx[b,c,iy,ix] = sum gy[b,k,oy,ox] w[k,c,r,s]
    with iy = 2*oy - pad + r.  w: [K, C, R, S] bf16."""
    return conv_transpose2d_subpixel(gy, w, None, pad, ih, iw, 0)


# -- autograd + module wrapper for decoder upconvs ----------------------
class _FusedDeconvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, act_code):
        y = deconv2d_fwd(x, w, bias, act_code)
        ctx.save_for_backward(x, w, y)
        ctx.meta = (act_code, bias is not None)
        return y

    @staticmethod
    def backward(ctx, gy):
        from .functional import require_hip

        x, w, y = ctx.saved_tensors
        act_code, has_bias = ctx.meta
        gy = gy.contiguous(memory_format=torch.channels_last)
        if act_code in (1, 2, 3):
            gy = require_hip().act_grad(gy, y, act_code)
        # dx of a transposed conv IS the plain stride-2 conv of gy with
        # the SAME [C_in, C_out, 4, 4] weight -> the MFMA fwd kernel
        gx = None
        need_gx = ctx.needs_input_grad[0]
        cout = w.shape[1]
        if need_gx and cout % 8 == 0 and (16 * cout) % 64 == 0:
            wcl = w.contiguous(memory_format=torch.channels_last)
            gx = require_hip().conv2d_fwd(gy, wcl, _empty().to(gy.device),
                                          2, 1, 0)
            need_gx = False
        need_gw = ctx.needs_input_grad[1]
        need_gb = has_bias and ctx.needs_input_grad[2]
        gx2 = gw = gb = None
        if need_gx or need_gw or need_gb:
            gx2, gw, gb = torch.ops.aten.convolution_backward(
                gy, x, w, [cout] if has_bias else None,
                [2, 2], [1, 1], [1, 1], True, [0, 0], 1,
                [need_gx, need_gw, need_gb])
        if gx is None:
            gx = gx2
        return gx, gw, gb, None


_deconv_cache: dict[tuple, str] = {}


class FusedDeconvAct(torch.nn.Module):
    """4x4/s2 transposed conv + bias + activation.

    GPU: sub-pixel MFMA kernel (4 parity launches, fused bias+act) when
    it measures faster than MIOpen for the shape; params live in a
    plain nn.ConvTranspose2d so bilinear init / checkpoints / warm
    starts are unchanged.
    """

    def __init__(self, cin: int, cout: int, act: str | None = "elu"):
        super().__init__()
        from .conv import _ACT_CODE

        self.deconv = torch.nn.ConvTranspose2d(cin, cout, 4, stride=2,
                                               padding=1, bias=True)
        self.act_name = act
        self.act_code = _ACT_CODE[act]
        self.cin, self.cout = cin, cout

    def _torch(self, x):
        from .conv import _act

        return _act(self.deconv(x), self.act_name)

    def _hip(self, x):
        w = self.deconv.weight
        if torch.is_autocast_enabled():
            x = x.to(torch.bfloat16)
            w = w.to(torch.bfloat16)
        if x.dtype != torch.bfloat16 or w.dtype != torch.bfloat16:
            return None
        x = x.contiguous(memory_format=torch.channels_last)
        return _FusedDeconvFn.apply(x, w, self.deconv.bias, self.act_code)

    def forward(self, x):
        import os

        if os.environ.get("DEEPOF_NO_DECONV"):
            return self._torch(x)
        if not (x.is_cuda and subpixel_eligible(self.cin, self.cout)
                and x.is_contiguous(memory_format=torch.channels_last)):
            return self._torch(x)
        key = ("deconv", self.cin, self.cout, tuple(x.shape), self.act_code)
        choice = _deconv_cache.get(key)
        if choice is None:
            choice = self._autotune(x, key)
        if choice == "hip":
            y = self._hip(x)
            if y is not None:
                return y
        return self._torch(x)

    @torch.no_grad()
    def _autotune(self, x, key) -> str:
        y = self._hip(x)
        if y is None:
            _deconv_cache[key] = "miopen"
            return "miopen"

        def timeit(fn, n=6):
            fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            return time.perf_counter() - t0

        try:
            t_h = timeit(lambda: self._hip(x))
            t_m = timeit(lambda: self._torch(x))
            choice = "hip" if t_h < t_m else "miopen"
        except Exception:
            choice = "miopen"
        _deconv_cache[key] = choice
        return choice


# -- measured dispatch (same policy as conv.py: time once per shape) ----
_bwd_cache: dict[tuple, str] = {}


def bwd_data_dispatch(gy: torch.Tensor, x_shape, w: torch.Tensor,
                      stride: int, pad: int):
    """Returns dx via the sub-pixel kernel when it measures faster than
    MIOpen's backward-data for this shape; None to fall back."""
    import os

    if os.environ.get("DEEPOF_NO_BWD2"):
        return None
    if stride != 2 or not subpixel_eligible(w.shape[0], w.shape[1]):
        return None
    ih, iw = x_shape[-2], x_shape[-1]
    key = ("bwd2", tuple(gy.shape), tuple(w.shape), pad)
    choice = _bwd_cache.get(key)
    if choice is None:
        def ours():
            return conv2d_bwd_data_subpixel(gy, w, pad, ih, iw)

        def mio():
            return torch.ops.aten.convolution_backward(
                gy, torch.empty(x_shape, device=gy.device, dtype=gy.dtype)
                .contiguous(memory_format=torch.channels_last),
                w, None, [stride, stride], [pad, pad], [1, 1], False,
                [0, 0], 1, [True, False, False])[0]

        def timeit(fn, n=4):
            fn()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            return time.perf_counter() - t0

        with torch.no_grad():
            try:
                t_h = timeit(ours)
                t_m = timeit(mio)
                choice = "hip" if t_h < t_m else "miopen"
            except Exception:
                choice = "miopen"
        _bwd_cache[key] = choice
    if choice == "hip":
        return conv2d_bwd_data_subpixel(gy, w, pad, ih, iw)
    return None
