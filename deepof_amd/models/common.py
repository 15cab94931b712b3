"""Shared building blocks: conv/deconv layers, bilinear deconv init,
and the 6-scale flow decoder used by every encoder family.

Reference parity: the decoder reproduces the expanding part of
/root/reference/flyingChairsWrapFlow.py:58-119 (per scale: 3x3 conv ->
2-channel flow head, 4x4/s2 deconv feature upsample with activation,
4x4/s2 linear deconv flow upsample, channel concat with the encoder
skip).  Activation is ELU (the reference's choice over LeakyReLU,
flyingChairsWrapFlow.py:29), selectable.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn


def make_act(name: str) -> nn.Module:
    if name == "elu":
        return nn.ELU(inplace=True)
    if name == "leaky_relu":
        return nn.LeakyReLU(0.1, inplace=True)
    if name == "relu":
        return nn.ReLU(inplace=True)
    raise ValueError(f"unknown activation {name!r}")


def conv(cin: int, cout: int, k: int = 3, stride: int = 1,
         act: str | None = "elu") -> nn.Module:
    from ..ops.conv import FusedConvAct

    return FusedConvAct(cin, cout, k, stride, act)


def deconv(cin: int, cout: int, act: str | None = "elu") -> nn.Module:
    """4x4 stride-2 transposed conv: exact 2x upsample (out = 2*in).

    On GPU the sub-pixel MFMA kernel runs it as 4 parity stride-1 convs
    with fused bias+act (ops/deconv.py); params live in the wrapped
    nn.ConvTranspose2d."""
    from ..ops.deconv import FusedDeconvAct

    return FusedDeconvAct(cin, cout, act)


def bilinear_deconv_weight(cin: int, cout: int, k: int = 4) -> torch.Tensor:
    """Bilinear-upsampling init for a stride-2 deconv weight [cin, cout, k, k].

    Matches the reference's deconv warm start
    (/root/reference/flyingChairsTrain.py:78-92): per-channel bilinear
    kernel on the diagonal, zeros elsewhere.
    """
    factor = (k + 1) // 2
    center = factor - 1 if k % 2 == 1 else factor - 0.5
    og = torch.arange(k, dtype=torch.float32)
    filt1d = 1 - torch.abs(og - center) / factor
    filt = torch.outer(filt1d, filt1d)
    w = torch.zeros(cin, cout, k, k)
    for i in range(min(cin, cout)):
        w[i, i] = filt
    return w


def init_flow_module(module: nn.Module) -> None:
    """Xavier conv init, zero bias; bilinear init for flow-upsample deconvs
    (modules flagged with ._bilinear_init = True)."""
    for m in module.modules():
        if isinstance(m, nn.Conv2d):
            nn.init.xavier_uniform_(m.weight)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.ConvTranspose2d):
            if getattr(m, "_bilinear_init", False):
                with torch.no_grad():
                    m.weight.copy_(
                        bilinear_deconv_weight(m.in_channels, m.out_channels,
                                               m.kernel_size[0])
                    )
            else:
                nn.init.xavier_uniform_(m.weight)
            if m.bias is not None:
                nn.init.zeros_(m.bias)


class FlowDecoder(nn.Module):
    """Multi-scale flow decoder.

    Args:
        feat_channels: encoder feature channels, coarsest first — e.g.
            FlowNetS: [1024, 512, 512, 256, 128, 64] for
            [conv6_2, conv5_2, conv4_2, conv3_2, conv2, conv1].
        up_channels: deconv output channels per refinement, e.g.
            [512, 256, 128, 64, 32].
        flow_channels: 2 for a single pair, 2*(T-1) for multi-frame
            Sintel volumes (sintelWrapFlow.py:386).

    forward(features) -> list of raw flow predictions, COARSEST first
    ([pr6 ... pr1] in reference naming).
    """

    def __init__(self, feat_channels: list[int], up_channels: list[int],
                 act: str = "elu", flow_channels: int = 2,
                 up_factors: list[int] | None = None):
        super().__init__()
        assert len(up_channels) == len(feat_channels) - 1
        self.num_scales = len(feat_channels)
        if up_factors is None:
            up_factors = [2] * (self.num_scales - 1)
        assert len(up_factors) == self.num_scales - 1
        self.flow_heads = nn.ModuleList()
        self.upconvs = nn.ModuleList()
        self.upflows = nn.ModuleList()

        def up(cin, cout, factor, a):
            # factor 1 happens between two same-resolution skips
            # (Inception's MaxPool_5a / Mixed_5d pair,
            # sintelWrapFlow.py:413-417 with scale=1)
            return deconv(cin, cout, act=a) if factor == 2 else conv(cin, cout, 3, 1, a)

        # concat channels padded to 64-multiples so the decoder's convs,
        # flow heads and upconvs are MFMA-kernel-eligible (1026/770/386/
        # 194/98 raw channels would force the MIOpen fallback everywhere)
        self._concat_pad: list[int] = [0] * self.num_scales
        concat_ch = feat_channels[0]
        for i in range(self.num_scales):
            self.flow_heads.append(conv(concat_ch, flow_channels, 3, act=None))
            if i < self.num_scales - 1:
                self.upconvs.append(up(concat_ch, up_channels[i], up_factors[i], act))
                uf = up(flow_channels, flow_channels, up_factors[i], None)
                from ..ops.deconv import FusedDeconvAct

                if isinstance(uf, nn.ConvTranspose2d):
                    uf._bilinear_init = True
                elif isinstance(uf, FusedDeconvAct):
                    uf.deconv._bilinear_init = True
                elif isinstance(uf, nn.Sequential):
                    uf[0]._bilinear_init = True
                else:  # FusedConvAct (stride-1 refinement stage)
                    uf.conv._bilinear_init = True
                self.upflows.append(uf)
                raw = feat_channels[i + 1] + up_channels[i] + flow_channels
                concat_ch = (raw + 63) // 64 * 64
                self._concat_pad[i + 1] = concat_ch - raw

    def _pad_zeros(self, skip, idx, h, w):
        """Cached read-only zero pad for the channel-64 alignment.

        The zeros MUST match the memory format of the other cat inputs:
        one NCHW tensor in a channels_last cat demotes the output to
        NCHW and every consumer then pays a layout-transposing copy
        (measured +10 ms/step on the r02 bench).  Cached per shape —
        the tensor is only ever READ (cat copies it), so reuse across
        steps is safe and removes a per-step alloc+fill.
        """
        fmt = (torch.channels_last
               if skip.is_contiguous(memory_format=torch.channels_last)
               else torch.contiguous_format)
        key = (idx, skip.shape[0], h, w, skip.dtype, str(skip.device), fmt)
        cache = getattr(self, "_pad_cache", None)
        if cache is None:
            cache = self._pad_cache = {}
        z = cache.get(key)
        if z is None:
            if len(cache) > 16:  # bound memory across batch-size changes
                cache.clear()
            z = torch.empty((skip.shape[0], self._concat_pad[idx], h, w),
                            dtype=skip.dtype, device=skip.device,
                            memory_format=fmt).zero_()
            cache[key] = z
        return z

    def forward(self, features: list[torch.Tensor]) -> list[torch.Tensor]:
        assert len(features) == self.num_scales
        flows = []
        x = features[0]
        for i in range(self.num_scales):
            pr = self.flow_heads[i](x)
            flows.append(pr)
            if i < self.num_scales - 1:
                skip = features[i + 1]
                up_feat = self.upconvs[i](x)
                up_flow = self.upflows[i](pr)
                # a 2x deconv of ceil(s/2) overshoots odd skips by 1 px
                h, w = skip.shape[-2:]
                parts = [skip, up_feat[..., :h, :w], up_flow[..., :h, :w]]
                if self._concat_pad[i + 1]:
                    parts.append(self._pad_zeros(skip, i + 1, h, w))
                x = torch.cat(parts, dim=1)
        return flows
