"""Sub-pixel (parity) decomposition of stride-2 transposed convs: the
host-side plan in ops/deconv.py must reproduce torch's conv_transpose2d
/ conv backward-data exactly (the GPU kernel consumes exactly this
plan; its numerics are covered by tests/test_gpu_kernels.py)."""

import math

import pytest
import torch
import torch.nn.functional as F

from deepof_amd.ops.deconv import _axis_plan


def subpixel_ref(x, w_nm, pad, out_h, out_w):
    """Emulate the 4 parity launches of conv2d_fwd_strided in torch."""
    B, M, H, W = x.shape
    N, _, R, S = w_nm.shape
    out = torch.zeros(B, N, out_h, out_w, dtype=x.dtype)
    for uy, taps_y, pad_y in _axis_plan(R, pad):
        for ux, taps_x, pad_x in _axis_plan(S, pad):
            if not taps_y or not taps_x or uy >= out_h or ux >= out_w:
                continue
            MH = math.ceil((out_h - uy) / 2)
            MW = math.ceil((out_w - ux) / 2)
            nry, nrx = len(taps_y), len(taps_x)
            # explicit zero pad so every (ty + j - pad_y) read is valid
            xp = F.pad(x, (pad_x, max(0, MW - 1 + nrx - 1 - pad_x - (W - 1)),
                           pad_y, max(0, MH - 1 + nry - 1 - pad_y - (H - 1))))
            sub = w_nm[:, :, taps_y][:, :, :, taps_x]
            y = F.conv2d(xp, sub)
            out[:, :, uy::2, ux::2] = y[:, :, :MH, :MW]
    return out


@pytest.mark.parametrize("R,pad", [(4, 1), (3, 1), (5, 2), (7, 3)])
def test_bwd_data_plan_matches_autograd(R, pad):
    """dx of a stride-2 conv via the parity plan == autograd's dx."""
    torch.manual_seed(0)
    B, C, K = 2, 8, 16
    IH, IW = 13, 17  # odd sizes stress the parity grids
    x = torch.randn(B, C, IH, IW, requires_grad=True)
    w = torch.randn(K, C, R, R)
    y = F.conv2d(x, w, stride=2, padding=pad)
    gy = torch.randn_like(y)
    (gx_ref,) = torch.autograd.grad(y, x, gy)

    gx = subpixel_ref(gy, w.transpose(0, 1), pad, IH, IW)
    assert torch.allclose(gx, gx_ref, atol=1e-4), \
        (gx - gx_ref).abs().max()


def test_deconv_plan_matches_conv_transpose():
    """4x4/s2 deconv forward via the parity plan == conv_transpose2d."""
    torch.manual_seed(1)
    B, C, K, H, W = 2, 8, 12, 9, 11
    x = torch.randn(B, C, H, W)
    w_ct = torch.randn(C, K, 4, 4)  # ConvTranspose2d weight layout
    ref = F.conv_transpose2d(x, w_ct, stride=2, padding=1)
    assert ref.shape[-2:] == (2 * H, 2 * W)
    got = subpixel_ref(x, w_ct.transpose(0, 1), 1, 2 * H, 2 * W)
    assert torch.allclose(got, ref, atol=1e-4)


def test_plan_covers_all_parities():
    for R, pad in [(4, 1), (3, 1), (5, 2), (7, 3)]:
        plans = _axis_plan(R, pad)
        offsets = sorted(u for u, taps, _ in plans if taps)
        assert offsets == [0, 1]
        total_taps = sum(len(t) for _, t, _ in plans)
        assert total_taps == R
        for _, taps, pad_p in plans:
            assert taps == sorted(taps, reverse=True)
            assert pad_p >= 0


def test_fused_deconv_module_cpu_fallback():
    """FusedDeconvAct == ConvTranspose2d + ELU on the CPU path, and the
    decoder's bilinear-init flag reaches the inner module."""
    from deepof_amd.models.common import FlowDecoder
    from deepof_amd.ops.deconv import FusedDeconvAct

    torch.manual_seed(2)
    m = FusedDeconvAct(8, 16, act="elu")
    x = torch.randn(2, 8, 6, 6)
    ref = F.elu(m.deconv(x))
    assert torch.allclose(m(x), ref)

    dec = FlowDecoder([64, 32], [16], act="elu", flow_channels=2)
    assert dec.upflows[0].deconv._bilinear_init
    feats = [torch.randn(1, 64, 4, 6), torch.randn(1, 32, 8, 12)]
    flows = dec(feats)
    assert flows[0].shape == (1, 2, 4, 6)
    assert flows[1].shape == (1, 2, 8, 12)


def test_fused_deconv_checkpoint_roundtrip(tmp_path):
    """FusedDeconvAct params live in the wrapped ConvTranspose2d:
    state_dict round-trips and bilinear init survives re-init."""
    import torch

    from deepof_amd.models import FlowNetS

    torch.manual_seed(0)
    m1 = FlowNetS()
    path = tmp_path / "m.pt"
    torch.save(m1.state_dict(), path)
    m2 = FlowNetS()
    missing, unexpected = m2.load_state_dict(
        torch.load(path, weights_only=True))
    assert not missing and not unexpected
    x = torch.randn(1, 6, 64, 96)
    f1 = m1(x)
    f2 = m2(x)
    for a, b in zip(f1, f2):
        assert torch.equal(a, b)


def test_decoder_pad_cache_reuse():
    """The cached pad-zeros tensor is reused across forwards (same
    object), stays zero, and respects batch-size changes."""
    import torch

    from deepof_amd.models.common import FlowDecoder

    dec = FlowDecoder([64, 32], [16], act="elu", flow_channels=2)
    assert dec._concat_pad[1] == 64 - (32 + 16 + 2)
    feats = [torch.randn(2, 64, 4, 6), torch.randn(2, 32, 8, 12)]
    dec(feats)
    (key1, z1), = dec._pad_cache.items()
    dec(feats)
    (key2, z2), = dec._pad_cache.items()
    assert key1 == key2 and z1 is z2
    assert z1.abs().sum() == 0
    feats4 = [torch.randn(4, 64, 4, 6), torch.randn(4, 32, 8, 12)]
    dec(feats4)
    assert len(dec._pad_cache) == 2  # both batch sizes cached
