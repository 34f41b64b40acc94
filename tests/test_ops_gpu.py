"""HIP kernel numerics vs plain PyTorch fp32 references (MI355X only).

Every kernel in ops/csrc is validated here against ops/reference.py in fp32
(and at bf16 I/O where the training path runs bf16)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from flreid_amd import ops
from flreid_amd.ops import reference as ref


@pytest.fixture(scope="module", autouse=True)
def _require_ext():
    assert torch.cuda.is_available()
    assert ops.extension_available(), "HIP extension must be built in-tree"


def test_l2norm_f32():
    x = torch.randn(37, 2048, device="cuda")
    y = ops.l2_normalize(x)
    assert torch.allclose(y, ref.l2_normalize(x), atol=1e-6)


def test_l2norm_bf16():
    x = torch.randn(16, 512, device="cuda").bfloat16()
    y = ops.l2_normalize(x)
    expected = ref.l2_normalize(x.float()).bfloat16()
    assert torch.allclose(y.float(), expected.float(), atol=2e-2)


def test_similarity_matrix_matches_mm():
    a = torch.randn(100, 512, device="cuda")
    b = torch.randn(300, 512, device="cuda")
    out = ops.similarity_matrix(a, b)
    expected = a @ b.t()
    assert torch.allclose(out, expected, atol=1e-4, rtol=1e-4)


def test_similarity_odd_sizes():
    # non-multiples of the 64x64 tile exercise the bounds guards
    a = torch.randn(65, 127, device="cuda")
    b = torch.randn(33, 127, device="cuda")
    out = ops.similarity_matrix(a, b)
    assert torch.allclose(out, a @ b.t(), atol=1e-4, rtol=1e-4)


def test_pairwise_sqeuclidean_gpu():
    a = torch.randn(70, 256, device="cuda")
    b = torch.randn(50, 256, device="cuda")
    out = ops.pairwise_sqeuclidean(a, b)
    expected = ref.pairwise_sqeuclidean(a, b)
    assert torch.allclose(out, expected, atol=1e-3, rtol=1e-4)


def test_pairwise_cosine_gpu():
    a = torch.randn(40, 128, device="cuda")
    b = torch.randn(60, 128, device="cuda")
    out = ops.pairwise_cosine_distance(a, b)
    expected = ref.pairwise_cosine_distance(a, b)
    assert torch.allclose(out, expected, atol=1e-5, rtol=1e-5)


def test_ce_smooth_fwd_bwd_f32():
    score = torch.randn(64, 8000, device="cuda", requires_grad=True)
    target = torch.randint(0, 8000, (64,), device="cuda")
    loss = ops.ce_label_smooth(score, target, 0.1)
    ref_score = score.detach().clone().requires_grad_(True)
    ref_loss = ref.ce_label_smooth(ref_score, target, 0.1)
    assert torch.allclose(loss, ref_loss, atol=1e-5, rtol=1e-5)
    loss.backward()
    ref_loss.backward()
    assert torch.allclose(score.grad, ref_score.grad, atol=1e-6)


def test_ce_smooth_bf16():
    score = torch.randn(32, 1000, device="cuda").bfloat16().requires_grad_(True)
    target = torch.randint(0, 1000, (32,), device="cuda")
    loss = ops.ce_label_smooth(score, target, 0.1)
    ref_loss = ref.ce_label_smooth(score.detach().float(), target, 0.1)
    assert abs(float(loss) - float(ref_loss)) < 0.05
    loss.backward()
    assert score.grad is not None and torch.isfinite(score.grad.float()).all()


def test_compose_fwd_bwd():
    gw = torch.randn(512, 512, device="cuda")
    atten = torch.full((512,), 0.9, device="cuda")
    aw = torch.randn(512, 512, device="cuda", requires_grad=True)
    out = ops.adaptive_compose(gw, atten, aw)
    expected = ref.adaptive_compose(gw, atten, aw.detach())
    assert torch.allclose(out, expected, atol=1e-6)
    out.sum().backward()
    assert torch.allclose(aw.grad, torch.ones_like(aw))


def test_compose_conv_shape_bf16():
    gw = torch.randn(128, 64, 3, 3, device="cuda").bfloat16()
    atten = torch.full((3,), 0.8, device="cuda")
    aw = torch.randn_like(gw)
    out = ops.adaptive_compose(gw, atten, aw)
    expected = ref.adaptive_compose(gw.float(), atten, aw.float())
    assert torch.allclose(out.float(), expected, atol=2e-2)


def test_importance_gpu():
    F = {"w": torch.zeros(1000, device="cuda")}
    g = {"w": torch.randn(1000, device="cuda")}
    ops.importance_update(F, g, mode="sq", scale=0.5)
    assert torch.allclose(F["w"], g["w"] ** 2 * 0.5, atol=1e-6)
    ops.importance_update(F, g, mode="abs", scale=2.0)
    assert torch.allclose(F["w"], g["w"] ** 2 * 0.5 + g["w"].abs() * 2.0,
                          atol=1e-5)


def test_cmc_map_gpu_matches_cpu():
    torch.manual_seed(0)
    qf = ref.l2_normalize(torch.randn(50, 256))
    gf = ref.l2_normalize(torch.randn(200, 256))
    ql = torch.randint(0, 20, (50,))
    gl = torch.randint(0, 20, (200,))
    cmc_cpu, map_cpu = ref.cmc_map(qf, ql, gf, gl)
    cmc_gpu, map_gpu = ops.cmc_map(qf.cuda(), ql, gf.cuda(), gl)
    assert torch.allclose(cmc_cpu, cmc_gpu, atol=1e-9)
    assert abs(map_cpu - map_gpu) < 1e-6


def test_window_attention_fused_vs_eager():
    torch.manual_seed(0)
    bw, h, n, d, nw = 32, 3, 49, 32, 4
    q = torch.randn(bw, h, n, d, device="cuda")
    k = torch.randn(bw, h, n, d, device="cuda")
    v = torch.randn(bw, h, n, d, device="cuda")
    bias = torch.randn(h, n, n, device="cuda")
    mask = torch.zeros(nw, n, n, device="cuda")
    mask[:, : n // 2, n // 2:] = -100.0
    with torch.no_grad():
        out = ops.window_attention(q, k, v, bias, mask, 0.17)
        expected = ref.window_attention(q, k, v, bias, mask, 0.17)
    assert torch.allclose(out, expected, atol=1e-4, rtol=1e-4)
    # no-mask path
    with torch.no_grad():
        out2 = ops.window_attention(q, k, v, bias, None, 0.17)
        exp2 = ref.window_attention(q, k, v, bias, None, 0.17)
    assert torch.allclose(out2, exp2, atol=1e-4, rtol=1e-4)


def test_window_attention_bf16():
    torch.manual_seed(1)
    q = torch.randn(16, 4, 49, 32, device="cuda").bfloat16()
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    bias = torch.randn(4, 49, 49, device="cuda")
    with torch.no_grad():
        out = ops.window_attention(q, k, v, bias, None, 0.2)
        expected = ref.window_attention(q.float(), k.float(), v.float(),
                                        bias, None, 0.2)
    assert torch.allclose(out.float(), expected, atol=3e-2, rtol=3e-2)


def test_triplet_fused_fwd_bwd_matches_eager():
    torch.manual_seed(0)
    feat = torch.randn(64, 512, device="cuda", requires_grad=True)
    target = torch.randint(0, 16, (64,), device="cuda")
    loss = ops.triplet_loss(feat, target, margin=0.3, hard_mining=True)
    ref_feat = feat.detach().clone().requires_grad_(True)
    ref_loss = ref.triplet_loss(ref_feat, target, margin=0.3, hard_mining=True)
    assert torch.allclose(loss, ref_loss, atol=1e-4, rtol=1e-4)
    (loss * 1.7).backward()
    (ref_loss * 1.7).backward()
    assert torch.allclose(feat.grad, ref_feat.grad, atol=1e-4, rtol=1e-4)


def test_triplet_fused_bf16_autocast_path():
    feat32 = torch.randn(32, 256, device="cuda", requires_grad=True)
    target = torch.randint(0, 8, (32,), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = ops.triplet_loss(feat32, target, margin=0.3, hard_mining=True)
    loss.backward()
    assert torch.isfinite(feat32.grad).all()


def test_cmc_map_chunked_matches_single():
    torch.manual_seed(2)
    qf = ref.l2_normalize(torch.randn(300, 128, device="cuda"))
    gf = ref.l2_normalize(torch.randn(700, 128, device="cuda"))
    ql = torch.randint(0, 40, (300,))
    gl = torch.randint(0, 40, (700,))
    cmc1, map1 = ops.cmc_map(qf, ql, gf, gl)
    cmc2, map2 = ops.cmc_map(qf, ql, gf, gl, query_chunk=64)
    assert torch.allclose(cmc1, cmc2, atol=1e-12)
    assert abs(map1 - map2) < 1e-9


def test_cmc_map_large_gallery():
    """iCaRL-scale eval (K8): 200k gallery through the MFMA GEMM + device
    ranking, chunked queries."""
    torch.manual_seed(3)
    g, q, d = 200_000, 256, 512
    gf = ref.l2_normalize(torch.randn(g, d, device="cuda"))
    qf = ref.l2_normalize(torch.randn(q, d, device="cuda"))
    gl = torch.randint(0, 5000, (g,))
    ql = gl[torch.randperm(g)[:q]]        # every query has matches
    cmc, mAP = ops.cmc_map(qf, ql, gf, gl, query_chunk=128)
    assert cmc.shape[0] == g
    assert 0.0 <= mAP <= 1.0
    assert float(cmc[-1]) <= 1.0 and float(cmc[0]) >= 0.0


def test_adaptive_linear_fused():
    """K2: fused compose-in-prologue GEMM vs eager compose + linear.
    Also pins down the gfx950 16x16x32 bf16 fragment k-layout."""
    torch.manual_seed(0)
    m, n, d = 64, 96, 128
    x = torch.randn(m, d, device="cuda").bfloat16()
    gw = torch.randn(n, d, device="cuda")
    aw = torch.randn(n, d, device="cuda") * 0.1
    atten = torch.rand(d, device="cuda")
    bias = torch.randn(n, device="cuda")
    expected = (x.float() @ (atten * gw + aw).t() + bias)
    got = None
    for layout in (1, 0):
        out = ops.adaptive_linear_fwd(x, gw, atten, aw, bias,
                                      split_layout=layout)
        if torch.allclose(out.float(), expected, atol=0.5, rtol=5e-2):
            got = layout
            break
    assert got is not None, "neither fragment layout matched the reference"
    # the shipped default must be the matching one
    out = ops.adaptive_linear_fwd(x, gw, atten, aw, bias)
    err = (out.float() - expected).abs().max()
    assert err < 0.5, f"default layout mismatch (max err {err}); matching={got}"


def test_adaptive_linear_fused_classifier_shape():
    m, n, d = 64, 8000, 2048
    x = torch.randn(m, d, device="cuda").bfloat16()
    gw = torch.randn(n, d, device="cuda") * 0.02
    aw = torch.randn(n, d, device="cuda") * 0.002
    atten = torch.full((d,), 0.9, device="cuda")
    out = ops.adaptive_linear_fwd(x, gw, atten, aw, None)
    expected = x.float() @ (atten * gw + aw).t()
    assert torch.allclose(out.float(), expected, atol=0.5, rtol=5e-2)


def test_adaptive_linear_autograd_matches_eager():
    torch.manual_seed(0)
    m, n, d = 32, 64, 128
    x = torch.randn(m, d, device="cuda", requires_grad=True)
    gw = torch.randn(n, d, device="cuda")
    aw = (torch.randn(n, d, device="cuda") * 0.1).requires_grad_(True)
    atten = torch.rand(d, device="cuda")
    bias = torch.zeros(n, device="cuda", requires_grad=True)
    out = ops.adaptive_linear(x, gw, atten, aw, bias)
    loss = (out.float() ** 2).mean()
    loss.backward()

    x2 = x.detach().clone().requires_grad_(True)
    aw2 = aw.detach().clone().requires_grad_(True)
    bias2 = bias.detach().clone().requires_grad_(True)
    theta = (atten * gw + aw2)
    out2 = torch.nn.functional.linear(x2.bfloat16(), theta.bfloat16(), bias2.bfloat16())
    loss2 = (out2.float() ** 2).mean()
    loss2.backward()
    assert torch.allclose(out.float(), out2.float(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(x.grad, x2.grad, atol=5e-3, rtol=5e-2)
    assert torch.allclose(aw.grad, aw2.grad, atol=5e-3, rtol=5e-2)
    assert torch.allclose(bias.grad, bias2.grad, atol=5e-3, rtol=5e-2)


def test_bn_eval_fused_matches_torch():
    import torch.nn as nn
    bn = nn.BatchNorm2d(32).cuda()
    bn.running_mean.uniform_(-1, 1)
    bn.running_var.uniform_(0.5, 2.0)
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.data.uniform_(-1, 1)
    bn.eval()
    for fmt in (torch.contiguous_format, torch.channels_last):
        x = torch.randn(8, 32, 16, 8, device="cuda").to(memory_format=fmt)
        out = ops.bn_eval_2d(x, bn)
        expected = bn(x)
        assert torch.allclose(out, expected, atol=1e-5, rtol=1e-5)
        xb = x.bfloat16()
        out_b = ops.bn_eval_2d(xb, bn)
        assert torch.allclose(out_b.float(), expected, atol=5e-2, rtol=5e-2)


def test_conv3x3_fwd_matches_torch():
    torch.manual_seed(0)
    for (n, c, h, w, k) in ((4, 32, 16, 8, 32), (2, 64, 7, 5, 16),
                            (8, 512, 16, 8, 512)):
        x = torch.randn(n, c, h, w, device="cuda").bfloat16() \
            .to(memory_format=torch.channels_last)
        wt = torch.randn(k, c, 3, 3, device="cuda") * (1.0 / c)
        out = ops.conv3x3_fwd_nhwc(x, wt)
        expected = torch.nn.functional.conv2d(x.float(), wt, padding=1)
        assert torch.allclose(out.float(), expected, atol=0.1, rtol=5e-2), \
            (n, c, h, w, k, (out.float() - expected).abs().max())


def test_bn_train_fused_matches_torch():
    """Fused train-mode BN (fwd stats+normalize+running update, analytic bwd)
    vs torch.nn.BatchNorm2d in fp32 (ops/csrc/bn_train.hip)."""
    import copy
    import torch.nn as nn
    torch.manual_seed(0)
    for dtype, atol in ((torch.float32, 1e-4), (torch.bfloat16, 5e-2)):
        for (n, c, h, w) in ((8, 64, 4, 4), (16, 128, 8, 4), (4, 2048, 8, 4)):
            bn = nn.BatchNorm2d(c).cuda()
            bn.weight.data.uniform_(0.5, 1.5)
            bn.bias.data.uniform_(-1, 1)
            bn.running_mean.uniform_(-1, 1)
            bn.running_var.uniform_(0.5, 2.0)
            bn_ref = copy.deepcopy(bn)
            bn.train(), bn_ref.train()

            x = (torch.randn(n, c, h, w, device="cuda")
                 .to(dtype).to(memory_format=torch.channels_last)
                 .requires_grad_(True))
            x_ref = x.detach().float().clone().requires_grad_(True)

            y = ops.bn_train_2d(x, bn)
            assert y is not None, "fused path did not engage"
            assert y.is_contiguous(memory_format=torch.channels_last)
            y_ref = bn_ref(x_ref)
            assert torch.allclose(y.float(), y_ref, atol=atol, rtol=5e-2)

            # running stats + counter updated like torch
            assert torch.allclose(bn.running_mean, bn_ref.running_mean,
                                  atol=atol, rtol=1e-2)
            assert torch.allclose(bn.running_var, bn_ref.running_var,
                                  atol=atol, rtol=1e-2)
            assert int(bn.num_batches_tracked) == int(bn_ref.num_batches_tracked)

            dy = torch.randn_like(y_ref)
            y.backward(dy.to(dtype))
            y_ref.backward(dy)
            assert torch.allclose(x.grad.float(), x_ref.grad,
                                  atol=atol * 10, rtol=5e-2), \
                (dtype, c, (x.grad.float() - x_ref.grad).abs().max())
            assert torch.allclose(bn.weight.grad, bn_ref.weight.grad,
                                  atol=atol * 10, rtol=5e-2)
            assert torch.allclose(bn.bias.grad, bn_ref.bias.grad,
                                  atol=atol * 10, rtol=5e-2)


def test_bn_train_fused_relu_matches_torch():
    """relu=True fuses BN→ReLU: fwd clamps, bwd masks dy by y>0
    (bn1/bn2 + stem inside the residual blocks)."""
    import copy
    import torch.nn as nn
    torch.manual_seed(3)
    bn = nn.BatchNorm2d(128).cuda().train()
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.data.uniform_(-1, 1)
    bn_ref = copy.deepcopy(bn)
    x = (torch.randn(16, 128, 8, 4, device="cuda")
         .to(memory_format=torch.channels_last).requires_grad_(True))
    x_ref = x.detach().clone().requires_grad_(True)
    y = ops.bn_train_2d(x, bn, relu=True)
    assert y is not None and (y >= 0).all()
    y_ref = torch.relu(bn_ref(x_ref))
    assert torch.allclose(y, y_ref, atol=1e-4, rtol=1e-3)
    dy = torch.randn_like(y_ref)
    y.backward(dy)
    y_ref.backward(dy)
    assert torch.allclose(x.grad, x_ref.grad, atol=1e-3, rtol=1e-2)
    assert torch.allclose(bn.weight.grad, bn_ref.weight.grad, atol=1e-3,
                          rtol=1e-2)
    assert torch.allclose(bn.bias.grad, bn_ref.bias.grad, atol=1e-3,
                          rtol=1e-2)

    # eval relu path
    bn.eval()
    xe = torch.randn(8, 128, 8, 4, device="cuda") \
        .to(memory_format=torch.channels_last)
    ye = ops.bn_eval_2d(xe, bn, relu=True)
    assert torch.allclose(ye, torch.relu(bn(xe)), atol=1e-5, rtol=1e-5)


def test_bn_train_fused_declines_out_of_regime():
    import torch.nn as nn
    bn = nn.BatchNorm2d(64).cuda().train()
    # NCHW-contiguous input -> decline
    x = torch.randn(8, 64, 4, 4, device="cuda")
    assert ops.bn_train_2d(x, bn) is None
    # big M (full-image training regime) -> decline
    xl = torch.randn(64, 64, 32, 16, device="cuda") \
        .to(memory_format=torch.channels_last)
    assert ops.bn_train_2d(xl, bn) is None
    # C % 64 != 0 -> decline
    bn2 = nn.BatchNorm2d(48).cuda().train()
    x2 = torch.randn(8, 48, 4, 4, device="cuda") \
        .to(memory_format=torch.channels_last)
    assert ops.bn_train_2d(x2, bn2) is None


def test_bn_train_fused_large_m_matches_torch():
    """The full-image fine-tune regime (fedavg/fedprox layer-4: M = 8192
    rows) through the fused train-BN — raised from the round-1 M<=4096
    cap."""
    import torch.nn as nn
    torch.manual_seed(9)
    bn = nn.BatchNorm2d(128).cuda().train()
    bn_ref = nn.BatchNorm2d(128).cuda().train()
    bn_ref.load_state_dict(bn.state_dict())
    x = (torch.randn(64, 128, 16, 8, device="cuda")
         .to(memory_format=torch.channels_last).requires_grad_(True))
    x2 = x.detach().clone().requires_grad_(True)
    y = ops.bn_train_2d(x, bn)
    assert y is not None
    y_ref = bn_ref(x2)
    assert torch.allclose(y, y_ref, atol=1e-4, rtol=1e-4)
    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(bn.weight.grad, bn_ref.weight.grad, atol=1e-3,
                          rtol=1e-3)
    assert torch.allclose(bn.running_mean, bn_ref.running_mean, atol=1e-5)
    assert torch.allclose(bn.running_var, bn_ref.running_var, atol=1e-4)


def test_bn_train_fused_1d_matches_torch():
    """2-D input path (BNNeck BatchNorm1d) of the fused train BN."""
    import copy
    import torch.nn as nn
    torch.manual_seed(1)
    bn = nn.BatchNorm1d(2048).cuda().train()
    bn.weight.data.uniform_(0.5, 1.5)
    bn.bias.requires_grad_(False)
    bn_ref = copy.deepcopy(bn)
    x = torch.randn(64, 2048, device="cuda", requires_grad=True)
    x_ref = x.detach().clone().requires_grad_(True)
    y = ops.bn_train_2d(x, bn)
    assert y is not None
    y_ref = bn_ref(x_ref)
    assert torch.allclose(y, y_ref, atol=1e-4, rtol=1e-3)
    assert torch.allclose(bn.running_var, bn_ref.running_var, atol=1e-4,
                          rtol=1e-3)
    assert int(bn.num_batches_tracked) == 1
    dy = torch.randn_like(y_ref)
    y.backward(dy)
    y_ref.backward(dy)
    assert torch.allclose(x.grad, x_ref.grad, atol=1e-3, rtol=1e-2)
    assert torch.allclose(bn.weight.grad, bn_ref.weight.grad, atol=1e-3,
                          rtol=1e-2)


def test_l1_drift_fused_matches_foreach():
    """Multi-tensor HIP drift (one-pass fwd, flat sign bwd) vs the _foreach
    reference (ops/csrc/drift.hip)."""
    torch.manual_seed(2)
    shapes = [(512, 512, 3, 3), (2048,), (8000, 2048), (7,)]
    params = [torch.randn(s, device="cuda", requires_grad=True)
              for s in shapes]
    anchors = [torch.randn(s, device="cuda") for s in shapes]
    pairs = [(p, a) for p, a in zip(params, anchors)]

    loss = ops.l1_drift(pairs)
    p2 = [p.detach().clone().requires_grad_(True) for p in params]
    ref_loss = ref.l1_drift_fused([(q, a) for q, a in zip(p2, anchors)])
    assert torch.allclose(loss, ref_loss, rtol=1e-6)

    loss.backward()
    ref_loss.backward()
    for p, q in zip(params, p2):
        assert torch.equal(p.grad, q.grad)

    # cached-metadata path: second call must reuse tables and stay correct
    loss2 = ops.l1_drift(pairs)
    assert torch.allclose(loss2, ref_loss, rtol=1e-6)


def test_adaptive_conv1x1_fused_route(monkeypatch):
    """Default-on fused pointwise route (compose2-to-bf16 + bf16 hipBLASLt,
    _AdaptiveLinear1x1Fn) vs the eager compose+linear fallback."""
    import copy
    from flreid_amd.models.adaptive import AdaptiveConv2d
    torch.manual_seed(4)
    w = torch.randn(512, 1024, 1, 1, device="cuda") * 0.02
    conv = AdaptiveConv2d(global_weight=w, stride=1, padding=0,
                          atten_default=0.9).cuda()
    conv2 = copy.deepcopy(conv)
    x = (torch.randn(4, 1024, 8, 4, device="cuda").bfloat16()
         .to(memory_format=torch.channels_last).requires_grad_(True))
    x2 = x.detach().clone().requires_grad_(True)

    # both routes run under bf16 autocast, like the training round
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = conv(x)                       # default: fused route
    monkeypatch.setenv("FLREID_NO_FUSED_1X1", "1")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y2 = conv2(x2)                    # eager compose + F.linear
    assert torch.allclose(y.float(), y2.float(), atol=0.1, rtol=5e-2)

    dy = torch.randn_like(y2.float())
    y.backward(dy.to(y.dtype))
    y2.backward(dy.to(y2.dtype))
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=0.1, rtol=5e-2)
    assert torch.allclose(conv.adaptive_weight.grad,
                          conv2.adaptive_weight.grad, atol=0.05, rtol=5e-2)


def test_kd_loss_fused_matches_reference():
    """K7 fused temperature-softmax KL (kd.hip) vs the fp32 eager
    reference (ref:criterions/kd_loss.py:10-27), loss AND grad."""
    torch.manual_seed(7)
    for T in (1.0, 4.0):
        zs = torch.randn(64, 8000, device="cuda", requires_grad=True)
        zt = torch.randn(64, 8000, device="cuda")
        loss = ops.kd_loss(zs, zt, T)
        loss.backward()
        zs_ref = zs.detach().clone().requires_grad_(True)
        expected = ref.kd_loss(zs_ref, zt, T)
        expected.backward()
        assert torch.allclose(loss, expected, atol=1e-4, rtol=1e-5)
        assert torch.allclose(zs.grad, zs_ref.grad, atol=1e-6)


def test_icarl_distill_fused_matches_reference():
    """K7 fused iCaRL distillation (both BCE-with-logits terms + combined
    grad, kd.hip) vs the eager two-loss composition
    (ref:methods/icarl.py:216-236)."""
    torch.manual_seed(11)
    B, C, P = 32, 96, 40
    score = torch.randn(B, C, device="cuda", requires_grad=True)
    target = torch.randint(0, C, (B,), device="cuda")
    prev = torch.randn(B, P, device="cuda")
    loss = ops.icarl_distill_loss(score, target, prev)
    loss.backward()

    score_ref = score.detach().clone().requires_grad_(True)
    expected = ref.icarl_distill_loss(score_ref, target, prev)
    expected.backward()
    assert torch.allclose(loss, expected, atol=1e-5, rtol=1e-5)
    assert torch.allclose(score.grad, score_ref.grad, atol=1e-7)


def _cl4(t):
    return t.to(memory_format=torch.channels_last)


@pytest.mark.parametrize("impl", ["auto", "hand"])
def test_conv3x3_img_fwd_matches_torch(impl, monkeypatch):
    """K1 conv path (conv3x3_img*.hip / fused-compose library route) fwd
    vs the fp32 library conv on bf16-rounded inputs."""
    monkeypatch.setenv("FLREID_CONV_IMPL", impl)
    torch.manual_seed(1)
    for (n, c, h, w, k) in ((4, 32, 16, 8, 64), (3, 64, 8, 4, 32),
                            (8, 512, 16, 8, 512), (2, 256, 16, 8, 512)):
        x = _cl4(torch.randn(n, c, h, w, device="cuda").bfloat16())
        wt = _cl4(torch.randn(k, c, 3, 3, device="cuda") * (1.0 / c))
        out = ops.conv3x3_try(x, wt)
        assert out is not None, (n, c, h, w, k)
        assert out.dtype == torch.bfloat16
        expected = torch.nn.functional.conv2d(
            x.float(), wt.float().to(memory_format=torch.contiguous_format),
            padding=1)
        assert torch.allclose(out.float(), expected, atol=0.1, rtol=5e-2), \
            (n, c, h, w, k, (out.float() - expected).abs().max())


@pytest.mark.parametrize("impl", ["auto", "hand"])
def test_conv3x3_img_plain_bwd_matches_torch(impl, monkeypatch):
    """dgrad (flip-transposed tile + the fwd kernel / conv2d_input) and
    wgrad (transpose-first MFMA reduction / conv2d_weight) vs torch
    autograd in fp32."""
    monkeypatch.setenv("FLREID_CONV_IMPL", impl)
    torch.manual_seed(2)
    n, c, h, w, k = 4, 64, 16, 8, 64
    x = _cl4(torch.randn(n, c, h, w, device="cuda").bfloat16())
    wt = _cl4(torch.randn(k, c, 3, 3, device="cuda") * (1.0 / c))
    x1 = x.clone().requires_grad_(True)
    w1 = wt.clone().requires_grad_(True)
    out = ops.conv3x3_try(x1, w1)
    g = torch.randn_like(out)
    out.backward(g)

    x2 = x.float().clone().requires_grad_(True)
    w2 = wt.clone().requires_grad_(True)
    ref_out = torch.nn.functional.conv2d(x2, w2, padding=1)
    ref_out.backward(g.float())

    assert w1.grad.dtype == torch.float32
    # wgrad reduces over 8192 rows of bf16 products in fp32
    assert torch.allclose(w1.grad, w2.grad, atol=0.5, rtol=5e-2), \
        (w1.grad - w2.grad).abs().max()
    assert torch.allclose(x1.grad.float(), x2.grad, atol=0.15, rtol=5e-2), \
        (x1.grad.float() - x2.grad).abs().max()


@pytest.mark.parametrize("impl", ["auto", "hand"])
def test_conv3x3_img_adaptive_fwd_bwd(impl, monkeypatch):
    """Composed path: θ = atten⊙gw + aw fused into the conv weight
    production; d(aw) arrives in fp32 (identity composition)."""
    monkeypatch.setenv("FLREID_CONV_IMPL", impl)
    torch.manual_seed(3)
    n, c, h, w, k = 2, 32, 16, 8, 32
    x = _cl4(torch.randn(n, c, h, w, device="cuda").bfloat16())
    gw = _cl4(torch.randn(k, c, 3, 3, device="cuda") * (1.0 / c))
    atten = torch.full((3,), 0.9, device="cuda")
    aw = _cl4((0.1 * gw).clone()).requires_grad_(True)
    out = ops.conv3x3_try(x, gw, atten, aw)
    assert out is not None
    g = torch.randn_like(out)
    out.backward(g)

    aw2 = aw.detach().clone().requires_grad_(True)
    theta = atten.view(1, 1, 1, 3) * gw + aw2
    ref = torch.nn.functional.conv2d(x.float(), theta, padding=1)
    ref.backward(g.float())
    assert torch.allclose(out.float(), ref.detach(), atol=0.1, rtol=5e-2)
    assert torch.allclose(aw.grad, aw2.grad, atol=0.3, rtol=5e-2), \
        (aw.grad - aw2.grad).abs().max()


def test_conv3x3_try_regime_guard():
    """Out-of-regime shapes must decline (caller falls back to library)."""
    x = _cl4(torch.randn(2, 64, 32, 16, device="cuda").bfloat16())  # HW=512
    wt = _cl4(torch.randn(64, 64, 3, 3, device="cuda"))
    assert ops.conv3x3_try(x, wt) is None
    x2 = _cl4(torch.randn(2, 24, 16, 8, device="cuda").bfloat16())  # C%32
    wt2 = _cl4(torch.randn(64, 24, 3, 3, device="cuda"))
    assert ops.conv3x3_try(x2, wt2) is None


def test_compose2_bf16_channels_last():
    """compose2: bf16-out composition in channels-last layout with the
    atten broadcast over the LOGICAL last dim (kernel width)."""
    from flreid_amd.ops import compose_theta_bf16, _load_extension
    ext = _load_extension()
    gw = _cl4(torch.randn(16, 32, 3, 3, device="cuda"))
    aw = _cl4(torch.randn(16, 32, 3, 3, device="cuda"))
    atten = torch.tensor([0.25, 0.5, 0.75], device="cuda")
    out = compose_theta_bf16(ext, gw, atten, aw)
    assert out.dtype == torch.bfloat16
    expected = (atten.view(1, 1, 1, 3) * gw + aw).bfloat16()
    assert torch.allclose(out.float(), expected.float(), atol=2e-2)


def test_window_attention_training_bwd_matches_eager():
    """K3 TRAINING path: fused fwd+bwd (window_attn.hip bwd kernel) vs the
    eager fp32 composition — dQ/dK/dV and the bias gradient."""
    torch.manual_seed(4)
    bw, h, n, d, nw = 8, 3, 49, 32, 4
    q = torch.randn(bw, h, n, d, device="cuda", requires_grad=True)
    k = torch.randn(bw, h, n, d, device="cuda", requires_grad=True)
    v = torch.randn(bw, h, n, d, device="cuda", requires_grad=True)
    bias = torch.randn(h, n, n, device="cuda", requires_grad=True)
    mask = (torch.randn(nw, n, n, device="cuda") > 0).float() * -100.0
    scale = d ** -0.5

    out = ops.window_attention(q, k, v, bias, mask, scale)
    g = torch.randn_like(out)
    out.backward(g)

    grads = [q.grad.clone(), k.grad.clone(), v.grad.clone(), bias.grad.clone()]
    for t in (q, k, v, bias):
        t.grad = None
    ref_out = ref.window_attention(q, k, v, bias, mask, scale, None)
    ref_out.backward(g)

    assert torch.allclose(out, ref_out, atol=1e-4, rtol=1e-4)
    for got, t, name in zip(grads, (q, k, v, bias), "qkvb"):
        assert torch.allclose(got, t.grad, atol=1e-3, rtol=1e-3), \
            (name, (got - t.grad).abs().max())


def test_window_attention_training_bwd_bf16():
    """bf16 I/O training path sanity (autocast regime)."""
    torch.manual_seed(5)
    bw, h, n, d = 4, 3, 49, 32
    q = torch.randn(bw, h, n, d, device="cuda").bfloat16().requires_grad_(True)
    k = torch.randn(bw, h, n, d, device="cuda").bfloat16().requires_grad_(True)
    v = torch.randn(bw, h, n, d, device="cuda").bfloat16().requires_grad_(True)
    bias = torch.randn(h, n, n, device="cuda", requires_grad=True)
    out = ops.window_attention(q, k, v, bias, None, d ** -0.5)
    out.sum().backward()

    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    bf = bias.detach().clone().requires_grad_(True)
    ref.window_attention(qf, kf, vf, bf, None, d ** -0.5, None).sum().backward()
    assert torch.allclose(q.grad.float(), qf.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(bias.grad, bf.grad, atol=5e-2, rtol=5e-2)


def test_patch_merge_ln_fwd_bwd_matches_eager():
    """K4 fused gather+LN (patch_merge.hip) vs the eager concat+LayerNorm
    composition in fp32 — output, dx, dgamma, dbeta."""
    torch.manual_seed(6)
    b, h, w, c = 3, 8, 4, 96
    x = torch.randn(b, h * w, c, device="cuda", requires_grad=True)
    gamma = torch.randn(4 * c, device="cuda", requires_grad=True)
    beta = torch.randn(4 * c, device="cuda", requires_grad=True)
    out = ops.patch_merge_ln(x, gamma, beta, h, w)
    assert out is not None and out.shape == (b, h * w // 4, 4 * c)
    g = torch.randn_like(out)
    out.backward(g)
    got = [x.grad.clone(), gamma.grad.clone(), beta.grad.clone()]
    for t in (x, gamma, beta):
        t.grad = None

    xv = x.view(b, h, w, c)
    cat = torch.cat([xv[:, 0::2, 0::2], xv[:, 1::2, 0::2],
                     xv[:, 0::2, 1::2], xv[:, 1::2, 1::2]], dim=-1)
    ref_out = torch.nn.functional.layer_norm(
        cat.reshape(b, -1, 4 * c), (4 * c,), gamma, beta)
    ref_out.backward(g)
    assert torch.allclose(out, ref_out, atol=1e-4, rtol=1e-4)
    for got_g, t, name in zip(got, (x, gamma, beta), ("dx", "dg", "db")):
        assert torch.allclose(got_g, t.grad, atol=1e-3, rtol=1e-3), \
            (name, (got_g - t.grad).abs().max())


def test_patch_merge_ln_bf16():
    torch.manual_seed(7)
    b, h, w, c = 2, 4, 4, 32
    x = torch.randn(b, h * w, c, device="cuda").bfloat16()
    gamma = torch.ones(4 * c, device="cuda")
    beta = torch.zeros(4 * c, device="cuda")
    out = ops.patch_merge_ln(x, gamma, beta, h, w)
    xv = x.float().view(b, h, w, c)
    cat = torch.cat([xv[:, 0::2, 0::2], xv[:, 1::2, 0::2],
                     xv[:, 0::2, 1::2], xv[:, 1::2, 1::2]], dim=-1)
    expected = torch.nn.functional.layer_norm(
        cat.reshape(b, -1, 4 * c), (4 * c,), gamma, beta)
    assert torch.allclose(out.float(), expected, atol=5e-2, rtol=5e-2)


def test_adaptive_conv3x3_production_dispatch():
    """The production AdaptiveConv2d 3×3 path must take the fused
    _Conv3x3Fn route (one-pass θ production + policy-routed conv): its
    output is bf16 channels-last and aw's gradient arrives in fp32 —
    the eager fallback under autocast would leave a plain autocast chain."""
    from flreid_amd.models.adaptive import AdaptiveConv2d
    torch.manual_seed(8)
    w = (torch.randn(64, 64, 3, 3, device="cuda") * 0.05) \
        .to(memory_format=torch.channels_last)
    conv = AdaptiveConv2d(global_weight=w, stride=1, padding=1,
                          atten_default=0.9).cuda()
    conv.adaptive_weight.data = conv.adaptive_weight.data \
        .to(memory_format=torch.channels_last)
    x = (torch.randn(4, 64, 16, 8, device="cuda").bfloat16()
         .to(memory_format=torch.channels_last))
    # the guard must accept exactly these production tensors
    direct = ops.conv3x3_try(x, conv.global_weight, conv.global_weight_atten,
                             conv.adaptive_weight)
    assert direct is not None, "fused conv route declined the production shape"
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = conv(x)
    assert y.dtype == torch.bfloat16
    assert y.is_contiguous(memory_format=torch.channels_last)
    assert torch.allclose(y.float(), direct.float(), atol=1e-2)
    y.float().sum().backward()
    assert conv.adaptive_weight.grad is not None
    assert conv.adaptive_weight.grad.dtype == torch.float32


def test_kd_loss_small_c_and_odd_shapes():
    """Strided row loops must handle C far below the block size and odd C."""
    torch.manual_seed(12)
    for (b, c) in ((3, 7), (5, 300), (1, 1)):
        zs = torch.randn(b, c, device="cuda", requires_grad=True)
        zt = torch.randn(b, c, device="cuda")
        loss = ops.kd_loss(zs, zt, 2.0)
        loss.backward()
        zs2 = zs.detach().clone().requires_grad_(True)
        expected = ref.kd_loss(zs2, zt, 2.0)
        expected.backward()
        assert torch.allclose(loss, expected, atol=1e-5, rtol=1e-5), (b, c)
        assert torch.allclose(zs.grad, zs2.grad, atol=1e-6), (b, c)


def test_patch_merge_ln_minimal_resolution():
    """H = W = 2 (one output token per image)."""
    torch.manual_seed(13)
    x = torch.randn(3, 4, 48, device="cuda", requires_grad=True)
    gamma = torch.randn(192, device="cuda", requires_grad=True)
    beta = torch.randn(192, device="cuda", requires_grad=True)
    out = ops.patch_merge_ln(x, gamma, beta, 2, 2)
    assert out.shape == (3, 1, 192)
    out.sum().backward()
    xv = x.detach().view(3, 2, 2, 48)
    cat = torch.cat([xv[:, 0::2, 0::2], xv[:, 1::2, 0::2],
                     xv[:, 0::2, 1::2], xv[:, 1::2, 1::2]], dim=-1)
    expected = torch.nn.functional.layer_norm(
        cat.reshape(3, 1, 192), (192,), gamma.detach(), beta.detach())
    assert torch.allclose(out.detach(), expected, atol=1e-4, rtol=1e-4)


def test_window_attention_bf16_masked_training():
    """bf16 + shift mask + autograd in one path (the shifted Swin blocks
    under autocast)."""
    torch.manual_seed(14)
    bw, h, n, d, nw = 8, 3, 49, 32, 4
    q = torch.randn(bw, h, n, d, device="cuda").bfloat16().requires_grad_(True)
    k = torch.randn_like(q).requires_grad_(True)
    v = torch.randn_like(q).requires_grad_(True)
    bias = torch.randn(h, n, n, device="cuda", requires_grad=True)
    mask = (torch.randn(nw, n, n, device="cuda") > 0.5).float() * -100.0
    out = ops.window_attention(q, k, v, bias, mask, d ** -0.5)
    out.sum().backward()
    for t, name in ((q, "q"), (k, "k"), (v, "v"), (bias, "bias")):
        assert t.grad is not None and torch.isfinite(t.grad.float()).all(), name

    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    bf = bias.detach().clone().requires_grad_(True)
    ref.window_attention(qf, kf, vf, bf, mask, d ** -0.5, None).sum().backward()
    assert torch.allclose(q.grad.float(), qf.grad, atol=6e-2, rtol=6e-2)
    assert torch.allclose(bias.grad, bf.grad, atol=6e-2, rtol=6e-2)


def test_compose2_odd_atten_length():
    """inner==1 with L % 4 != 0 exercises the scalar tail path."""
    from flreid_amd.ops import compose_theta_bf16, _load_extension
    ext = _load_extension()
    gw = torch.randn(10, 6, device="cuda")
    aw = torch.randn_like(gw)
    atten = torch.rand(6, device="cuda")
    out = compose_theta_bf16(ext, gw, atten, aw)
    expected = (atten * gw + aw).bfloat16()
    assert torch.allclose(out.float(), expected.float(), atol=2e-2)
