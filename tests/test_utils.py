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


def test_flo_error_paths(tmp_path):
    import pytest

    p = tmp_path / "bad.flo"
    p.write_bytes(b"\x00" * 12)
    with pytest.raises(ValueError, match="magic"):
        read_flo(p)
    # truncated payload
    import struct

    p2 = tmp_path / "trunc.flo"
    p2.write_bytes(struct.pack("<fii", 202021.25, 4, 4) + b"\x00" * 8)
    with pytest.raises(ValueError, match="truncated"):
        read_flo(p2)
    with pytest.raises(ValueError, match="H, W, 2"):
        write_flo(tmp_path / "x.flo", np.zeros((4, 4, 3), dtype=np.float32))


def test_config_roundtrip(tmp_path):
    import yaml

    from deepof_amd.config import Config

    cfg = Config(dataset="sintel", batch_size=7, lambda_smooth=0.0,
                 image_size=(256, 512))
    p = tmp_path / "c.yaml"
    p.write_text(yaml.safe_dump(cfg.to_dict()))
    back = Config.from_yaml(str(p))
    assert back.to_dict() == cfg.to_dict()
    # overrides + unknown-key rejection
    over = back.apply_overrides(["lr=0.001", "model=flownetc"])
    assert over.lr == 0.001 and over.model == "flownetc"
    import pytest

    with pytest.raises(ValueError, match="unknown config key"):
        back.apply_overrides(["nonsense=1"])
