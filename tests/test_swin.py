"""Swin backbone (ref:models/swin_transformer.py surface)."""

import torch

from flreid_amd.models import nets


def test_swin_tiny_dual_output_small_input():
    # small img_size keeps the CPU test fast; window math identical
    net = nets["swin_transformer_tiny"](num_classes=16, neck="bnneck",
                                        img_size=32, window_size=4)
    x = torch.randn(2, 3, 64, 32)    # resized internally to img_size²
    net.train()
    score, feat = net(x)
    assert score.shape == (2, 16)
    assert feat.shape == (2, net.in_planes)
    net.eval()
    assert net(x).shape == (2, net.in_planes)


def test_swin_shifted_mask_built():
    net = nets["swin_transformer_tiny"](num_classes=8, neck="no", img_size=32, window_size=4)
    blk = net.base.layers[0].blocks[1]      # odd block -> shifted
    assert blk.shift > 0
    assert blk.attn_mask is not None


def test_swin_staged_head_matches_full():
    net = nets["swin_transformer_tiny"](num_classes=8, neck="bnneck",
                                        img_size=32, window_size=4)
    net.eval()
    x = torch.randn(1, 3, 32, 32)
    with torch.no_grad():
        full, tap = net.run_stages(x, 0, tap=4)       # tap input of layers.3
        head_out, _ = net.run_stages(tap, start=4)
    assert torch.allclose(full, head_out, atol=1e-5)


def test_swin_fedstil_conversion():
    from flreid_amd.runtime.builder import parser_model
    model = parser_model("fedstil", {
        "name": "swin_transformer_tiny", "num_classes": 16, "neck": "bnneck",
        "img_size": 32, "window_size": 4, "atten_default": 0.9, "lambda_l1": 1e-3,
        "lambda_k": 32, "fine_tuning": ["base.layers.3", "classifier"]})
    assert model.head_stage == 4
    x = torch.randn(2, 3, 32, 32)
    model.eval()
    with torch.no_grad():
        _out, tap = model.tap_forward(x)
        feat = model.head_forward(tap)
    assert feat.shape == (2, model.net.in_planes)
