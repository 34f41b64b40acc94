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
