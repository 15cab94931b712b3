"""Golden-value regression tests: fixed seeds -> exact expected losses.

Guards the loss semantics (warp taps, masks, normalizers, Charbonnier
exponents) against silent refactor drift; the HIP kernels are tested
against the same reference in test_gpu_kernels.py, so these anchor the
whole chain.
"""

import torch

from deepof_amd.ops import reference as ref


def _fixture():
    torch.manual_seed(42)
    img1 = torch.rand(2, 3, 32, 48)
    img2 = torch.rand(2, 3, 32, 48)
    flow = torch.randn(2, 2, 32, 48) * 0.3
    return flow, img1, img2


def test_unsup_loss_golden():
    flow, img1, img2 = _fixture()
    res = ref.unsup_loss_scale(flow, img1, img2, flow_scale=5.0,
                               epsilon=1e-4, alpha_c=0.25, alpha_s=0.37,
                               lambda_smooth=1.0)
    assert abs(float(res["total"]) - 10.835532) < 2e-4
    assert abs(float(res["photo"]) - 8.038655) < 2e-4
    assert abs(float(res["u_loss"]) - 1.383160) < 2e-4
    assert abs(float(res["v_loss"]) - 1.413717) < 2e-4


def test_unsup_loss_golden_sintel_hypers():
    # Sintel config: alpha_c = alpha_s = 0.3, lambda = 0
    flow, img1, img2 = _fixture()
    res = ref.unsup_loss_scale(flow, img1, img2, flow_scale=2.5,
                               epsilon=1e-4, alpha_c=0.3, alpha_s=0.3,
                               lambda_smooth=0.0)
    # lambda 0: total == photo
    torch.testing.assert_close(res["total"], res["photo"])
    assert torch.isfinite(res["total"])


def test_border_mask_area_golden():
    # 384x512 (the flagship config): bw = ceil(38.4) = 39
    m = ref.border_mask(384, 512)
    assert float(m.sum()) == (384 - 78) * (512 - 78)
