"""CPU-fallback op semantics vs plain torch (these same semantics are what
the GPU kernels are tested against in tests/test_kernels_gpu.py)."""

import pytest
import torch
import torch.nn.functional as F

from active_learning_amd.ops import functional as AF
from active_learning_amd.ops.loss import cross_entropy
from active_learning_amd.ops.optim import FusedAdam, FusedSGD


def _nhwc(x_nchw):
    return x_nchw.permute(0, 2, 3, 1).contiguous()


def test_conv2d_matches_torch_fwd_bwd():
    x = torch.randn(2, 8, 9, 5, requires_grad=True)          # NHWC
    w = torch.randn(6, 3, 3, 5, requires_grad=True)          # KRSC
    y = AF.conv2d(x, w, stride=2, padding=1)
    x_t = x.detach().permute(0, 3, 1, 2).requires_grad_(True)
    w_t = w.detach().permute(0, 3, 1, 2).requires_grad_(True)
    y_t = F.conv2d(x_t, w_t, stride=2, padding=1)
    assert torch.allclose(y, _nhwc(y_t), atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    y_t.backward(dy.permute(0, 3, 1, 2))
    assert torch.allclose(x.grad, _nhwc(x_t.grad), atol=1e-5)
    assert torch.allclose(w.grad, w_t.grad.permute(0, 2, 3, 1), atol=1e-5)


@pytest.mark.parametrize("training", [True, False])
@pytest.mark.parametrize("relu", [True, False])
def test_batch_norm_act_matches_torch(training, relu):
    n, h, w, c = 4, 5, 6, 7
    x = torch.randn(n, h, w, c, requires_grad=True)
    gamma = torch.randn(c, requires_grad=True).abs() + 0.5
    gamma = gamma.detach().requires_grad_(True)
    beta = torch.randn(c, requires_grad=True)
    rm = torch.randn(c)
    rv = torch.rand(c) + 0.5

    bn = torch.nn.BatchNorm2d(c)
    with torch.no_grad():
        bn.weight.copy_(gamma)
        bn.bias.copy_(beta)
        bn.running_mean.copy_(rm)
        bn.running_var.copy_(rv)
    bn.train(training)

    rm2, rv2 = rm.clone(), rv.clone()
    y = AF.batch_norm_act(x, gamma, beta, rm2, rv2, use_batch_stats=training,
                          relu=relu)
    x_t = x.detach().permute(0, 3, 1, 2).requires_grad_(True)
    y_t = bn(x_t)
    if relu:
        y_t = F.relu(y_t)
    assert torch.allclose(y, _nhwc(y_t), atol=1e-5)
    # running-stat updates (training mode)
    if training:
        assert torch.allclose(rm2, bn.running_mean, atol=1e-5)
        assert torch.allclose(rv2, bn.running_var, atol=1e-4)
    dy = torch.randn_like(y)
    y.backward(dy)
    y_t.backward(dy.permute(0, 3, 1, 2))
    assert torch.allclose(x.grad, _nhwc(x_t.grad), atol=1e-4)
    assert torch.allclose(gamma.grad, bn.weight.grad, atol=1e-4)
    assert torch.allclose(beta.grad, bn.bias.grad, atol=1e-4)


def test_bn_residual_fusion():
    n, h, w, c = 2, 4, 4, 3
    x = torch.randn(n, h, w, c, requires_grad=True)
    res = torch.randn(n, h, w, c, requires_grad=True)
    gamma = torch.ones(c, requires_grad=True)
    beta = torch.zeros(c, requires_grad=True)
    rm, rv = torch.zeros(c), torch.ones(c)
    y = AF.batch_norm_act(x, gamma, beta, rm, rv, use_batch_stats=True, relu=True,
                          residual=res)
    # unfused reference
    x2 = x.detach().requires_grad_(True)
    r2 = res.detach().requires_grad_(True)
    xf = x2.float()
    mean = xf.mean(dim=(0, 1, 2))
    var = xf.var(dim=(0, 1, 2), unbiased=False)
    y2 = F.relu((xf - mean) / (var + 1e-5).sqrt() + r2)
    assert torch.allclose(y, y2, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    y2.backward(dy)
    assert torch.allclose(res.grad, r2.grad, atol=1e-5)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)


def test_maxpool_matches_torch():
    x = torch.randn(2, 9, 9, 4, requires_grad=True)
    y = AF.max_pool2d(x, 3, 2, 1)
    x_t = x.detach().permute(0, 3, 1, 2).requires_grad_(True)
    y_t = F.max_pool2d(x_t, 3, 2, 1)
    assert torch.allclose(y, _nhwc(y_t), atol=1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)
    y_t.backward(dy.permute(0, 3, 1, 2))
    assert torch.allclose(x.grad, _nhwc(x_t.grad), atol=1e-6)


def test_global_avg_pool():
    x = torch.randn(3, 4, 5, 6, requires_grad=True)
    y = AF.global_avg_pool(x)
    expected = x.detach().mean(dim=(1, 2))
    assert torch.allclose(y, expected, atol=1e-6)
    y.sum().backward()
    assert torch.allclose(x.grad, torch.full_like(x, 1.0 / 20))


@pytest.mark.parametrize("weighted", [False, True])
def test_cross_entropy_matches_torch(weighted):
    logits = torch.randn(16, 10, requires_grad=True)
    targets = torch.randint(0, 10, (16,))
    w = torch.rand(10) + 0.1 if weighted else None
    loss = cross_entropy(logits, targets, w)
    l_t = logits.detach().requires_grad_(True)
    expected = F.cross_entropy(l_t, targets, weight=w)
    assert torch.allclose(loss, expected, atol=1e-6)
    loss.backward()
    expected.backward()
    assert torch.allclose(logits.grad, l_t.grad, atol=1e-6)


def test_fused_sgd_matches_torch():
    p1 = torch.nn.Parameter(torch.randn(13))
    p2 = torch.nn.Parameter(torch.randn(13))
    with torch.no_grad():
        p2.copy_(p1)
    opt1 = FusedSGD([p1], lr=0.1, momentum=0.9, weight_decay=1e-2)
    opt2 = torch.optim.SGD([p2], lr=0.1, momentum=0.9, weight_decay=1e-2)
    for _ in range(5):
        g = torch.randn(13)
        p1.grad = g.clone()
        p2.grad = g.clone()
        opt1.step()
        opt2.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_fused_adam_matches_torch():
    p1 = torch.nn.Parameter(torch.randn(17))
    p2 = torch.nn.Parameter(p1.detach().clone())
    opt1 = FusedAdam([p1], lr=1e-2, weight_decay=1e-3)
    opt2 = torch.optim.Adam([p2], lr=1e-2, weight_decay=1e-3)
    for _ in range(5):
        g = torch.randn(17)
        p1.grad = g.clone()
        p2.grad = g.clone()
        opt1.step()
        opt2.step()
    assert torch.allclose(p1, p2, atol=1e-5)


def test_conv_transpose_matches_torch():
    x = torch.randn(2, 8, 8, 5, requires_grad=True)
    w = torch.randn(5, 4, 4, 3, requires_grad=True)  # (Cin, R, S, Cout)
    from active_learning_amd.models.layers import _TransposedConv2d
    y = _TransposedConv2d.apply(x, w, 2, 1)
    x_t = x.detach().permute(0, 3, 1, 2).requires_grad_(True)
    w_t = w.detach().permute(0, 3, 1, 2).requires_grad_(True)
    y_t = F.conv_transpose2d(x_t, w_t, stride=2, padding=1)
    assert torch.allclose(y, _nhwc(y_t), atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    y_t.backward(dy.permute(0, 3, 1, 2))
    assert torch.allclose(x.grad, _nhwc(x_t.grad), atol=1e-5)
    assert torch.allclose(w.grad, w_t.grad.permute(0, 2, 3, 1), atol=1e-4)
