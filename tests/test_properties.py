"""Property-based tests (hypothesis): .flo round-trip, metrics
invariants, sub-pixel plan tables, config override round-trip."""

import numpy as np
import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from deepof_amd.utils import read_flo, write_flo
from deepof_amd.utils.metrics import endpoint_error


@settings(max_examples=25, deadline=None)
@given(h=st.integers(1, 40), w=st.integers(1, 40),
       seed=st.integers(0, 2**31 - 1))
def test_flo_roundtrip(tmp_path_factory, h, w, seed):
    rng = np.random.default_rng(seed)
    flow = rng.standard_normal((h, w, 2)).astype(np.float32) * 50
    p = tmp_path_factory.mktemp("flo") / "f.flo"
    write_flo(p, flow)
    back = read_flo(p)
    assert back.shape == (h, w, 2)
    assert np.array_equal(back, flow)  # bit-exact round trip


@settings(max_examples=25, deadline=None)
@given(h=st.integers(2, 24), w=st.integers(3, 24),
       seed=st.integers(0, 2**31 - 1), mag=st.floats(0.0, 100.0))
def test_epe_invariants(h, w, seed, mag):
    # w >= 3: a trailing dim of exactly 2 is read as an HWC uv layout
    # by the metric's layout inference (reference .flo convention)
    g = torch.Generator().manual_seed(seed)
    f = torch.randn(1, 2, h, w, generator=g)
    # identity -> zero error
    assert float(endpoint_error(f, f)) == 0.0
    # constant offset of magnitude m in u -> EPE exactly m
    off = f.clone()
    off[:, 0] += mag
    assert abs(float(endpoint_error(off, f)) - mag) < 1e-3 * max(mag, 1)
    # symmetry
    other = torch.randn(1, 2, h, w, generator=g)
    assert abs(float(endpoint_error(f, other))
               - float(endpoint_error(other, f))) < 1e-5


@settings(max_examples=40, deadline=None)
@given(r=st.integers(1, 8), s=st.integers(1, 8), pad_raw=st.integers(0, 8))
def test_plan_tabs_wellformed(r, s, pad_raw):
    """The device tables of the single-launch sub-pixel path must tile
    the weight exactly: per-parity blocks are disjoint, cover all
    N*M*R*S elements, and tap indices are valid/descending."""
    from deepof_amd.ops.deconv import _axis_plan

    pad = pad_raw % (min(r, s) // 2 + 1)  # realistic: 2*pad < R
    ys = _axis_plan(r, pad)
    xs = _axis_plan(s, pad)
    # each axis: offsets {0,1} among non-empty parities, taps partition,
    # and the plan reproduces the transposed-conv scatter map exactly:
    # out(oy) = sum x(t) w(r) over {(t, r): 2t - pad + r == oy}
    for R_, plans in ((r, ys), (s, xs)):
        taps_all = sorted(t for _, taps, _ in plans for t in taps)
        assert taps_all == list(range(R_))
        for u, taps, pad_p in plans:
            assert u in (0, 1)
            assert taps == sorted(taps, reverse=True)
            for ty in range(6):  # a few output positions of this parity
                oy = 2 * ty + u
                got = sorted((ty + j - pad_p, taps[j])
                             for j in range(len(taps)))
                ref = sorted((t, rr) for t in range(-8, 16)
                             for rr in range(R_) if 2 * t - pad + rr == oy)
                assert got == ref, (R_, pad, u, oy)
    n_elems = sum(len(ty) * len(tx)
                  for _, ty, _ in ys for _, tx, _ in xs)
    assert n_elems == r * s


@settings(max_examples=20, deadline=None)
@given(lr=st.floats(1e-7, 1.0), bs=st.integers(1, 512),
       lam=st.floats(0.0, 10.0))
def test_config_override_roundtrip(lr, bs, lam):
    from deepof_amd.config import Config

    cfg = Config().apply_overrides(
        [f"lr={lr!r}", f"batch_size={bs}", f"lambda_smooth={lam!r}"])
    assert cfg.lr == lr and cfg.batch_size == bs
    assert cfg.lambda_smooth == lam
    # round-trip through dict preserves everything
    assert Config.from_dict(cfg.to_dict()) == cfg


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(0, 2**31 - 1))
def test_photometric_augment_range(seed):
    """Photometric augmentation must keep images in [0, 255] and apply
    the SAME transform to both frames of a pair (the loss warps one
    onto the other)."""
    from deepof_amd.utils.augment import photometric_augment

    g = torch.Generator().manual_seed(seed)
    img1 = torch.rand(2, 3, 16, 20, generator=g) * 255
    img2 = torch.rand(2, 3, 16, 20, generator=g) * 255
    a1, a2 = photometric_augment(img1, img2, generator=g)
    for a in (a1, a2):
        assert a.shape == img1.shape
        assert float(a.min()) >= 0.0 and float(a.max()) <= 255.0
    # identical inputs -> identical outputs up to the additive noise
    b1, b2 = photometric_augment(img1, img1,
                                 generator=torch.Generator().manual_seed(7),
                                 noise_sigma=0.0)
    assert torch.allclose(b1, b2)


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(0, 2**31 - 1))
def test_geometric_augment_shape_and_shared_transform(seed):
    from deepof_amd.utils.augment import geometric_augment

    g = torch.Generator().manual_seed(seed)
    img = torch.rand(2, 3, 16, 20, generator=g) * 255
    a1, a2 = geometric_augment(img, img,
                               generator=torch.Generator().manual_seed(3))
    assert a1.shape == img.shape
    # same input + same transform -> both outputs identical
    assert torch.allclose(a1, a2)


def test_fused_adam_cpu_matches_torch_adam():
    """The CPU foreach path (the numerics reference for the HIP kernel)
    must match torch.optim.Adam exactly (no weight decay: torch's Adam
    uses decoupled grad add like ours)."""
    from deepof_amd.engine.optim import FusedAdam

    torch.manual_seed(0)
    shapes = [(64,), (8, 8), (3, 4, 5)]
    pa = [torch.randn(s, requires_grad=True) for s in shapes]
    pb = [p.detach().clone().requires_grad_(True) for p in pa]
    grads = [torch.randn(s) for s in shapes]
    for p, g in zip(pa, grads):
        p.grad = g.clone()
    for p, g in zip(pb, grads):
        p.grad = g.clone()
    oa = FusedAdam(pa, lr=1e-2, betas=(0.9, 0.999), eps=1e-8)
    ob = torch.optim.Adam(pb, lr=1e-2, betas=(0.9, 0.999), eps=1e-8)
    for _ in range(5):
        oa.step()
        ob.step()
    for a, b in zip(pa, pb):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-8)


def test_read_flo_rejects_bad_magic(tmp_path_factory):
    from deepof_amd.utils import read_flo

    p = tmp_path_factory.mktemp("bad") / "x.flo"
    p.write_bytes(b"\x00\x00\x00\x00" + b"\x01\x00\x00\x00" * 2 + b"\x00" * 8)
    with pytest.raises(Exception):
        read_flo(p)

