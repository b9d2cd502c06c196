"""CPU reference-path correctness: our autograd ops vs plain torch.

These CPU implementations are the numerics references the HIP kernels are
tested against on the GPU (tests/test_gpu_ops.py), so they must themselves
match torch.
"""

import torch
import torch.nn.functional as F

from pytorch_ddp_template_amd.ops import functional as X


def assert_close(a, b, rtol=1e-5, atol=1e-5):
    torch.testing.assert_close(a, b, rtol=rtol, atol=atol)


def test_linear_matches_torch():
    x = torch.randn(8, 10, requires_grad=True)
    w = torch.randn(5, 10, requires_grad=True)
    b = torch.randn(5, requires_grad=True)
    y = X.linear(x, w, b)
    y.sum().backward()
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = F.linear(xr, wr, br)
    yr.sum().backward()
    assert_close(y, yr)
    assert_close(x.grad, xr.grad)
    assert_close(w.grad, wr.grad)
    assert_close(b.grad, br.grad)


def test_linear_fused_relu():
    x = torch.randn(8, 10, requires_grad=True)
    w = torch.randn(5, 10, requires_grad=True)
    y = X.linear(x, w, None, act="relu")
    y.sum().backward()
    xr = x.detach().clone().requires_grad_(True)
    yr = torch.relu(F.linear(xr, w.detach()))
    yr.sum().backward()
    assert_close(y, yr)
    assert_close(x.grad, xr.grad)


def test_conv2d_nhwc_matches_torch():
    x = torch.randn(2, 8, 8, 3, requires_grad=True)
    w = torch.randn(6, 3, 3, 3, requires_grad=True)  # (K, R, S, C)
    y = X.conv2d_nhwc(x, w, None, stride=2, pad=1)
    y.float().pow(2).sum().backward()
    xr = x.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    wr = w.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, 2, 1)
    yr.pow(2).sum().backward()
    assert_close(y.permute(0, 3, 1, 2), yr, rtol=1e-4, atol=1e-4)
    assert_close(x.grad, xr.grad.permute(0, 2, 3, 1), rtol=1e-4, atol=1e-4)
    assert_close(w.grad, wr.grad.permute(0, 2, 3, 1), rtol=1e-4, atol=1e-4)


def test_batchnorm_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(4, 5, 5, 7, requires_grad=True)
    gamma = torch.randn(7, requires_grad=True)
    beta = torch.randn(7, requires_grad=True)
    rm = torch.zeros(7)
    rv = torch.ones(7)
    y = X.batch_norm2d_nhwc(x, gamma, beta, rm, rv, training=True, momentum=0.1)
    y.pow(2).sum().backward()

    xr = x.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    gr = gamma.detach().clone().requires_grad_(True)
    br = beta.detach().clone().requires_grad_(True)
    rm2 = torch.zeros(7)
    rv2 = torch.ones(7)
    yr = F.batch_norm(xr, rm2, rv2, gr, br, training=True, momentum=0.1)
    yr.pow(2).sum().backward()
    assert_close(y.permute(0, 3, 1, 2), yr, rtol=1e-4, atol=1e-4)
    assert_close(x.grad, xr.grad.permute(0, 2, 3, 1), rtol=1e-4, atol=1e-4)
    assert_close(gamma.grad, gr.grad, rtol=1e-4, atol=1e-4)
    assert_close(beta.grad, br.grad, rtol=1e-4, atol=1e-4)
    assert_close(rm, rm2, rtol=1e-5, atol=1e-6)
    assert_close(rv, rv2, rtol=1e-5, atol=1e-6)


def test_cross_entropy_matches_torch():
    logits = torch.randn(16, 10, requires_grad=True)
    target = torch.randint(0, 10, (16,))
    loss = X.cross_entropy(logits, target)
    loss.backward()
    lr = logits.detach().clone().requires_grad_(True)
    ref = F.cross_entropy(lr, target)
    ref.backward()
    assert_close(loss, ref)
    assert_close(logits.grad, lr.grad)


def test_mse_matches_torch():
    p = torch.randn(8, 5, requires_grad=True)
    t = torch.randn(8, 5)
    loss = X.mse_loss(p, t)
    loss.backward()
    pr = p.detach().clone().requires_grad_(True)
    ref = F.mse_loss(pr, t)
    ref.backward()
    assert_close(loss, ref)
    assert_close(p.grad, pr.grad)


def test_layernorm_matches_torch():
    x = torch.randn(6, 32, requires_grad=True)
    g = torch.randn(32, requires_grad=True)
    b = torch.randn(32, requires_grad=True)
    y = X.layer_norm(x, g, b, eps=1e-6)
    y.pow(2).sum().backward()
    xr = x.detach().clone().requires_grad_(True)
    gr = g.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = F.layer_norm(xr, (32,), gr, br, eps=1e-6)
    yr.pow(2).sum().backward()
    assert_close(y, yr, rtol=1e-4, atol=1e-5)
    assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-4)
    assert_close(g.grad, gr.grad, rtol=1e-4, atol=1e-4)
    assert_close(b.grad, br.grad, rtol=1e-4, atol=1e-4)


def test_softmax_attention_matches_torch():
    q = torch.randn(3, 7, 16, requires_grad=True)
    k = torch.randn(3, 7, 16, requires_grad=True)
    v = torch.randn(3, 7, 16, requires_grad=True)
    out = X.attention(q, k, v, 0.25)
    out.sum().backward()
    qr, kr, vr = (t.detach().clone().requires_grad_(True) for t in (q, k, v))
    ref = torch.softmax(qr @ kr.transpose(1, 2) * 0.25, dim=-1) @ vr
    ref.sum().backward()
    assert_close(out, ref, rtol=1e-4, atol=1e-5)
    assert_close(q.grad, qr.grad, rtol=1e-4, atol=1e-4)
    assert_close(k.grad, kr.grad, rtol=1e-4, atol=1e-4)
    assert_close(v.grad, vr.grad, rtol=1e-4, atol=1e-4)


def test_maxpool_and_avgpool():
    x = torch.randn(2, 8, 8, 4, requires_grad=True)
    y = X.max_pool2d_nhwc(x, 3, 2, 1)
    y.sum().backward()
    xr = x.detach().permute(0, 3, 1, 2).clone().requires_grad_(True)
    yr = F.max_pool2d(xr, 3, 2, 1)
    yr.sum().backward()
    assert_close(y.permute(0, 3, 1, 2), yr)
    assert_close(x.grad, xr.grad.permute(0, 2, 3, 1))

    x2 = torch.randn(2, 5, 5, 3, requires_grad=True)
    p = X.global_avg_pool_nhwc(x2)
    p.sum().backward()
    assert_close(p, x2.detach().mean(dim=(1, 2)))
    assert_close(x2.grad, torch.full_like(x2, 1.0 / 25))


def test_add_relu_and_gelu():
    a = torch.randn(32, requires_grad=True)
    b = torch.randn(32, requires_grad=True)
    y = X.add_relu(a, b)
    y.sum().backward()
    mask = (a.detach() + b.detach() > 0).float()
    assert_close(y, torch.relu(a.detach() + b.detach()))
    assert_close(a.grad, mask)
    assert_close(b.grad, mask)

    x = torch.randn(64, requires_grad=True)
    g = X.gelu(x)
    g.sum().backward()
    xr = x.detach().clone().requires_grad_(True)
    gr = F.gelu(xr)
    gr.sum().backward()
    assert_close(g, gr, rtol=1e-5, atol=1e-6)
    assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-5)
