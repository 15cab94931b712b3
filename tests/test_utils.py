import numpy as np
import torch

from deepof_amd.utils import (angular_error, endpoint_error, flow_to_color,
                              read_flo, write_flo)


def test_flo_roundtrip(tmp_path):
    flow = np.random.default_rng(0).standard_normal((17, 23, 2)).astype(np.float32)
    p = tmp_path / "x.flo"
    write_flo(p, flow)
    back = read_flo(p)
    assert back.shape == (17, 23, 2)
    np.testing.assert_array_equal(back, flow)


def test_flo_reads_reference_layout(tmp_path):
    # byte-level check: magic, w, h, interleaved u,v row-major
    p = tmp_path / "y.flo"
    flow = np.zeros((2, 3, 2), dtype=np.float32)
    flow[0, 1] = (1.5, -2.5)
    write_flo(p, flow)
    raw = np.fromfile(p, dtype=np.float32)
    assert raw[0] == np.float32(202021.25)
    assert raw.view(np.int32)[1] == 3 and raw.view(np.int32)[2] == 2
    assert raw[3 + 2] == 1.5 and raw[3 + 3] == -2.5


def test_endpoint_error():
    f1 = torch.zeros(1, 4, 5, 2)
    f2 = torch.zeros(1, 4, 5, 2)
    f2[..., 0] = 3.0
    f2[..., 1] = 4.0
    assert abs(endpoint_error(f1, f2) - 5.0) < 1e-6
    # NCHW layout accepted too
    assert abs(endpoint_error(f1.permute(0, 3, 1, 2), f2.permute(0, 3, 1, 2)) - 5.0) < 1e-6


def test_angular_error_zero():
    f = torch.ones(1, 4, 5, 2)
    assert angular_error(f, f) < 1e-6


def test_flow_to_color():
    rng = np.random.default_rng(0)
    flow = rng.standard_normal((8, 9, 2)).astype(np.float32)
    img = flow_to_color(flow)
    assert img.shape == (8, 9, 3) and img.dtype == np.uint8
    # zero flow -> near-white (saturation ~0 at radius 0)
    white = flow_to_color(np.zeros((4, 4, 2), dtype=np.float32))
    assert (white > 200).all()
