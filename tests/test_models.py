"""Backbone tests (ref:models/resnet.py surface)."""

import torch

from flreid_amd.models import nets


def test_resnet18_dual_output():
    net = nets["resnet18"](num_classes=32, last_stride=1, neck="bnneck")
    x = torch.randn(2, 3, 64, 32)
    net.train()
    score, feat = net(x)
    assert score.shape == (2, 32)
    assert feat.shape == (2, 512)
    net.eval()
    feat = net(x)
    assert feat.shape == (2, 512)


def test_resnet50_bottleneck_channels():
    net = nets["resnet50"](num_classes=8, last_stride=1, neck="bnneck")
    net.eval()
    feat = net(torch.randn(1, 3, 64, 32))
    assert feat.shape == (1, 2048)


def test_last_stride_keeps_resolution():
    net1 = nets["resnet18"](num_classes=8, last_stride=1, neck="no")
    net2 = nets["resnet18"](num_classes=8, last_stride=2, neck="no")
    x = torch.randn(1, 3, 64, 32)
    f1 = net1.base.layer4(net1.base.layer3(net1.base.layer2(net1.base.layer1(
        net1.base.maxpool(net1.base.relu(net1.base.bn1(net1.base.conv1(x))))))))
    f2 = net2.base.layer4(net2.base.layer3(net2.base.layer2(net2.base.layer1(
        net2.base.maxpool(net2.base.relu(net2.base.bn1(net2.base.conv1(x))))))))
    assert f1.shape[-1] == 2 * f2.shape[-1]


def test_fine_tuning_freeze():
    from flreid_amd.runtime.builder import parser_model
    model = parser_model("fedavg", {
        "name": "resnet18", "num_classes": 16, "last_stride": 1,
        "neck": "bnneck", "fine_tuning": ["base.layer4", "classifier"]})
    frozen = [n for n, p in model.net.named_parameters() if not p.requires_grad]
    live = [n for n, p in model.net.named_parameters() if p.requires_grad]
    assert any(n.startswith("base.layer1") for n in frozen)
    assert all(n.startswith(("base.layer4", "classifier")) for n in live)
    assert any(n.startswith("base.layer4") for n in live)


def test_bnneck_classifier_has_no_bias():
    net = nets["resnet18"](num_classes=8, neck="bnneck")
    assert net.classifier.bias is None
    assert net.bottleneck.bias.requires_grad is False


def test_staged_execution_matches_full_forward():
    """run_stages with a tap must (a) reproduce forward() exactly and
    (b) let a head-only invocation on the tap reproduce the full output —
    the fx-free FedSTIL split (models/resnet.py:STAGES)."""
    import torch
    from flreid_amd.models.resnet import resnet18

    torch.manual_seed(0)
    m = resnet18(num_classes=8, neck="bnneck", last_stride=1)
    m.eval()
    x = torch.randn(2, 3, 64, 32)

    full = m(x)
    out, tap = m.run_stages(x, start=0, tap=4)     # tap = layer4 input
    assert torch.allclose(out, full, atol=1e-6)
    assert tap is not None and tap.dim() == 4

    head_out, _ = m.run_stages(tap, start=4)
    assert torch.allclose(head_out, full, atol=1e-6)


def test_stage_of_mapping():
    from flreid_amd.models.resnet import resnet18

    m = resnet18(num_classes=8, neck="bnneck")
    assert m.stage_of("base.conv1") == 0
    assert m.stage_of("base.layer1.0.conv1") == 1
    assert m.stage_of("base.layer4.1.bn2") == 4
    assert m.stage_of("classifier") == len(m.STAGES) - 1
    assert m.stage_of("bottleneck") == len(m.STAGES) - 1
