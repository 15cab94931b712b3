import pytest
import torch

from deepof_amd.models import FlowNetC, FlowNetS, InceptionFlow, VGG16Flow
from deepof_amd.models.common import bilinear_deconv_weight
from deepof_amd.models.flownet import FLOW_SCALES


def test_flownets_shapes():
    m = FlowNetS()
    x = torch.randn(2, 6, 96, 128)
    flows = m(x)
    assert len(flows) == 6
    # finest first: pr1 at 1/2 .. pr6 at 1/64
    h, w = 96, 128
    for k, f in enumerate(flows):
        # strided convs use ceil division per level
        eh, ew = h, w
        for _ in range(k + 1):
            eh, ew = -(-eh // 2), -(-ew // 2)
        assert f.shape == (2, 2, eh, ew), (k, f.shape)


def test_flownets_param_count():
    m = FlowNetS()
    n = sum(p.numel() for p in m.parameters())
    # reference prints ~38M for FlowNetS (flyingChairsTrain.py:118)
    assert 30e6 < n < 50e6, n


def test_flow_scales():
    assert FLOW_SCALES == [10.0, 5.0, 2.5, 1.25, 0.625, 0.3125]


def test_flownetc_shapes():
    m = FlowNetC(max_displacement=2)  # small md for CPU speed
    x = torch.randn(1, 6, 64, 96)
    flows = m(x)
    assert len(flows) == 6
    assert flows[0].shape == (1, 2, 32, 48)
    assert flows[5].shape == (1, 2, 1, 1) or flows[5].shape[-1] >= 1


def test_vgg16_shapes():
    m = VGG16Flow()
    x = torch.randn(1, 6, 64, 96)
    flows = m(x)
    assert len(flows) == 5
    for k, f in enumerate(flows):
        s = 1 << (k + 1)
        assert f.shape == (1, 2, 64 // s, 96 // s)


def test_inception_shapes():
    m = InceptionFlow()
    x = torch.randn(1, 6, 128, 192)
    flows = m(x)
    assert len(flows) == 6
    # finest at 1/2; two 1/8 scales in the middle
    assert flows[0].shape == (1, 2, 64, 96)
    assert flows[2].shape[-2:] == flows[3].shape[-2:]  # both 1/8


def test_inception_multiframe():
    m = InceptionFlow(time_step=4)
    x = torch.randn(1, 12, 64, 96)
    flows = m(x)
    assert flows[0].shape[1] == 6  # 2*(T-1)


def test_bilinear_deconv_weight():
    w = bilinear_deconv_weight(2, 2, 4)
    assert w.shape == (2, 2, 4, 4)
    # off-diagonal channels zero; kernel rows sum to the bilinear profile
    assert w[0, 1].abs().sum() == 0
    torch.testing.assert_close(w[0, 0], w[1, 1])
    # upsampling a constant field by this kernel preserves the constant
    x = torch.ones(1, 2, 5, 5)
    import torch.nn.functional as F

    y = F.conv_transpose2d(x, w, stride=2, padding=1)
    torch.testing.assert_close(y[..., 2:-2, 2:-2],
                               torch.ones_like(y[..., 2:-2, 2:-2]))


def test_backward_runs():
    m = FlowNetS()
    x = torch.randn(1, 6, 64, 64)
    flows = m(x)
    loss = sum(f.abs().mean() for f in flows)
    loss.backward()
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert len(grads) > 0
    assert all(torch.isfinite(g).all() for g in grads)
