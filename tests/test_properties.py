"""Property-based tests (hypothesis): .flo round-trip, metrics
invariants, sub-pixel plan tables, config override round-trip."""

import numpy as np
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
