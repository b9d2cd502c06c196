"""SGD / clip / schedule parity with torch (CPU reference path)."""

import math

import torch

from pytorch_ddp_template_amd.optim import (
    SGD,
    clip_grad_norm_,
    get_linear_schedule_with_warmup,
)


def _pair(seed=0, **kw):
    torch.manual_seed(seed)
    p1 = [torch.randn(20, 10, requires_grad=True), torch.randn(10, requires_grad=True)]
    p2 = [p.detach().clone().requires_grad_(True) for p in p1]
    g = [torch.randn_like(p) for p in p1]
    for a, b, gg in zip(p1, p2, g):
        a.grad = gg.clone()
        b.grad = gg.clone()
    return p1, p2


def test_sgd_plain_matches_torch():
    p1, p2 = _pair()
    ours = SGD(p1, lr=0.1)
    ref = torch.optim.SGD(p2, lr=0.1)
    for _ in range(3):
        ours.step()
        ref.step()
    for a, b in zip(p1, p2):
        torch.testing.assert_close(a, b, rtol=1e-6, atol=1e-7)


def test_sgd_momentum_wd_matches_torch():
    p1, p2 = _pair(1)
    ours = SGD(p1, lr=0.05, momentum=0.9, weight_decay=1e-4)
    ref = torch.optim.SGD(p2, lr=0.05, momentum=0.9, weight_decay=1e-4)
    for _ in range(4):
        ours.step()
        ref.step()
    for a, b in zip(p1, p2):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_sgd_nesterov_matches_torch():
    p1, p2 = _pair(2)
    ours = SGD(p1, lr=0.05, momentum=0.9, nesterov=True)
    ref = torch.optim.SGD(p2, lr=0.05, momentum=0.9, nesterov=True)
    for _ in range(3):
        ours.step()
        ref.step()
    for a, b in zip(p1, p2):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_sgd_master_weights_bf16():
    torch.manual_seed(3)
    p32 = torch.randn(50)
    p = p32.to(torch.bfloat16).requires_grad_(True)
    p.grad = torch.randn(50).to(torch.bfloat16)
    opt = SGD([p], lr=0.1, master_weights=True)
    opt.step()
    master = opt.state[p]["master"]
    # master updated in fp32 from the bf16 starting point
    expected = p32.to(torch.bfloat16).float() - 0.1 * p.grad.float()
    torch.testing.assert_close(master, expected, rtol=1e-6, atol=1e-7)
    torch.testing.assert_close(p.detach(), expected.to(torch.bfloat16))


def test_clip_grad_norm_matches_torch():
    torch.manual_seed(4)
    p1 = [torch.randn(30, requires_grad=True) for _ in range(3)]
    p2 = [p.detach().clone().requires_grad_(True) for p in p1]
    for a, b in zip(p1, p2):
        g = torch.randn_like(a) * 10
        a.grad = g.clone()
        b.grad = g.clone()
    n1 = clip_grad_norm_(p1, 1.0)
    n2 = torch.nn.utils.clip_grad_norm_(p2, 1.0)
    torch.testing.assert_close(n1, n2, rtol=1e-5, atol=1e-6)
    for a, b in zip(p1, p2):
        torch.testing.assert_close(a.grad, b.grad, rtol=1e-5, atol=1e-6)


def test_linear_warmup_schedule_shape():
    p = [torch.zeros(1, requires_grad=True)]
    opt = SGD(p, lr=1.0)
    sched = get_linear_schedule_with_warmup(opt, 10, 100)
    lrs = []
    for _ in range(100):
        lrs.append(sched.get_last_lr()[0])
        opt.step()
        sched.step()
    assert math.isclose(lrs[0], 0.0)
    assert math.isclose(lrs[5], 0.5)
    assert math.isclose(lrs[10], 1.0)
    assert lrs[50] < 1.0
    assert math.isclose(lrs[99], (100 - 99) / 90, rel_tol=1e-6)


def test_sgd_state_dict_roundtrip_keeps_fp32_state():
    """torch's Optimizer.load_state_dict casts state to the param dtype —
    with bf16 params that rounded the fp32 master/momentum through bf16
    (and crashed the fused GPU kernel on the first post-resume step).  Our
    override must restore the fp32 slots bit-exactly from the saved
    tensors."""
    import torch

    from pytorch_ddp_template_amd.optim import SGD

    p = torch.nn.Parameter(torch.randn(64).to(torch.bfloat16))
    opt = SGD([p], lr=0.1, momentum=0.9, master_weights=True)
    p.grad = torch.randn_like(p)
    opt.step()
    sd = opt.state_dict()
    # fresh optimizer, same param object
    opt2 = SGD([p], lr=0.1, momentum=0.9, master_weights=True)
    opt2.load_state_dict(sd)
    st = opt2.state[p]
    assert st["momentum_buffer"].dtype == torch.float32
    assert st["master"].dtype == torch.float32
    assert torch.equal(st["master"], opt.state[p]["master"])
    assert torch.equal(st["momentum_buffer"], opt.state[p]["momentum_buffer"])
    p.grad = torch.randn_like(p)
    opt2.step()  # must not crash / degrade
