"""End-to-end model paths on MI355X: HIP kernels vs fp32 CPU references,
loss descent, and the native extension actually being used."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_native_extension_loaded():
    from pytorch_ddp_template_amd.ops import native_available

    assert native_available(), "HIP extension must load on the GPU box"


def test_foo_model_matches_cpu_f32():
    from pytorch_ddp_template_amd.models import FooModel

    torch.manual_seed(0)
    m = FooModel()
    mg = FooModel()
    mg.load_state_dict(m.state_dict())
    mg = mg.to(DEV)
    x = torch.randn(64, 10)
    y = m(x)
    yg = mg(x.to(DEV))
    torch.testing.assert_close(yg.cpu(), y, rtol=1e-3, atol=1e-4)
    # grads
    y.pow(2).sum().backward()
    yg.pow(2).sum().backward()
    for p, pg in zip(m.parameters(), mg.parameters()):
        torch.testing.assert_close(pg.grad.cpu(), p.grad, rtol=2e-2, atol=2e-3)


def test_resnet18_bf16_forward_close_to_cpu_f32():
    from pytorch_ddp_template_amd.models import resnet18

    torch.manual_seed(1)
    m = resnet18(num_classes=10, stem="cifar")
    m.eval()
    x = torch.randn(4, 32, 32, 3)
    with torch.no_grad():
        ref = m(x)
        mg = resnet18(num_classes=10, stem="cifar")
        mg.load_state_dict(m.state_dict())
        mg.eval()
        out = mg.to(torch.bfloat16).to(DEV)(x.to(torch.bfloat16).to(DEV))
    err = (out.float().cpu() - ref).abs().max()
    scale = ref.abs().max().clamp(min=1.0)
    assert err / scale < 0.12, f"relative error {err/scale}"


def test_resnet18_bf16_training_loss_descends():
    from pytorch_ddp_template_amd.models import resnet18
    from pytorch_ddp_template_amd.ops import CrossEntropyLoss
    from pytorch_ddp_template_amd.optim import SGD

    torch.manual_seed(2)
    m = resnet18(num_classes=10, stem="cifar").to(torch.bfloat16).to(DEV)
    opt = SGD(m.parameters(), lr=0.05, momentum=0.9, master_weights=True)
    crit = CrossEntropyLoss()
    x = torch.randn(64, 32, 32, 3).to(torch.bfloat16).to(DEV)
    y = torch.randint(0, 10, (64,)).to(DEV)
    losses = []
    for _ in range(20):
        out = m(x)
        loss = crit(out, y)
        loss.backward()
        opt.step()
        m.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.7, losses[:3] + losses[-3:]
    assert all(l == l for l in losses), "NaN loss"


def test_resnet18_training_grads_finite_under_allocator_churn():
    """Regression for the round-2 wide-N bug (see test_gpu_ops.py
    test_conv2d_fwd_stats_wide_n_shape): steady-state training loops recycle
    the same allocator blocks each step, so a kernel that leaves part of its
    output unwritten reads stale-but-plausible values and loss still
    descends.  Alternate batch sizes to churn the allocator and assert every
    grad/param stays finite — this fails loudly on unwritten-memory bugs."""
    from pytorch_ddp_template_amd.models import resnet18
    from pytorch_ddp_template_amd.ops import CrossEntropyLoss
    from pytorch_ddp_template_amd.optim import SGD, clip_grad_norm_

    torch.manual_seed(7)
    m = resnet18(num_classes=10, stem="cifar").to(torch.bfloat16).to(DEV)
    opt = SGD(m.parameters(), lr=0.01, momentum=0.9, master_weights=True)
    crit = CrossEntropyLoss()
    for i, bs in enumerate([256, 192, 320, 256, 64, 256]):
        x = torch.randn(bs, 32, 32, 3).to(torch.bfloat16).to(DEV)
        y = torch.randint(0, 10, (bs,)).to(DEV)
        loss = crit(m(x), y)
        loss.backward()
        for n, p in m.named_parameters():
            assert torch.isfinite(p.grad.float()).all(), f"step {i}: {n} grad"
        clip_grad_norm_(list(m.parameters()), 1000.0)
        opt.step()
        m.zero_grad(set_to_none=False)
        torch.cuda.synchronize()
        for n, p in m.named_parameters():
            assert torch.isfinite(p.float()).all(), f"step {i}: {n} param"
        assert float(loss) < 6.0, f"step {i}: loss {float(loss)}"


def test_foo_training_with_mse_descends():
    from pytorch_ddp_template_amd.models import FooModel
    from pytorch_ddp_template_amd.ops import MSELoss
    from pytorch_ddp_template_amd.optim import SGD, clip_grad_norm_

    torch.manual_seed(3)
    m = FooModel().to(DEV)
    opt = SGD(m.parameters(), lr=0.05)
    crit = MSELoss()
    x = torch.randn(256, 10).to(DEV)
    # learnable target (random y's best loss is ~Var(y): nothing to descend)
    w_true = torch.randn(10, 5).to(DEV) * 0.5
    y = x @ w_true
    losses = []
    for _ in range(80):
        loss = crit(m(x), y)
        loss.backward()
        clip_grad_norm_(list(m.parameters()), 1000.0)
        opt.step()
        m.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.5, (losses[0], losses[-1])


def test_vit_tiny_bf16_fwd_bwd_runs():
    from pytorch_ddp_template_amd.models.vit import ViT

    torch.manual_seed(4)
    m = (
        ViT(image_size=32, patch_size=8, dim=64, depth=2, heads=4, num_classes=10)
        .to(torch.bfloat16)
        .to(DEV)
    )
    x = torch.randn(4, 32, 32, 3).to(torch.bfloat16).to(DEV)
    y = m(x)
    assert y.shape == (4, 10)
    y.float().sum().backward()
    assert m.blocks[0].attn.qkv.weight.grad is not None
    assert torch.isfinite(y.float()).all()


def test_resnet50_bf16_fwd_bwd_runs():
    from pytorch_ddp_template_amd.models import resnet50

    m = resnet50(num_classes=1000).to(torch.bfloat16).to(DEV)
    x = torch.randn(2, 224, 224, 3).to(torch.bfloat16).to(DEV)
    y = m(x)
    assert y.shape == (2, 1000)
    y.float().sum().backward()
    assert torch.isfinite(y.float()).all()


def test_checkpoint_roundtrip_gpu(tmp_path):
    from pytorch_ddp_template_amd.ddp import save_model
    from pytorch_ddp_template_amd.models import resnet18

    m = resnet18().to(torch.bfloat16).to(DEV)
    save_model(m, str(tmp_path / "ck"))
    m2 = resnet18().to(torch.bfloat16).to(DEV)
    sd = torch.load(tmp_path / "ck" / "model.bin", map_location="cpu",
                    weights_only=True)
    m2.load_state_dict(sd)
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a.cpu(), b.cpu())


@pytest.mark.gpu
def test_fused_basic_block_matches_unfused():
    """The fused single-node BasicBlock path (conv-epilogue BN stats, fused
    ReLU masks, in-epilogue residual add) must match the unfused module
    composition — forward, input grad, and every param grad."""
    from pytorch_ddp_template_amd.models.resnet import BasicBlock

    torch.manual_seed(0)
    for in_ch, ch, stride in [(64, 64, 1), (64, 128, 2)]:
        blk = BasicBlock(in_ch, ch, stride).to(torch.bfloat16).cuda()
        blk.train()
        x = torch.randn(16, 8, 8, in_ch, dtype=torch.bfloat16, device="cuda")
        x1 = x.clone().requires_grad_(True)
        x2 = x.clone().requires_grad_(True)

        out_f = blk(x1)          # fused (training + native + pow2)
        loss_f = out_f.float().square().mean()
        loss_f.backward()
        gf = [p.grad.clone() for p in blk.parameters()]
        xf = x1.grad.clone()
        rm_f = blk.conv1[1].running_mean.clone()

        # reset state, force the unfused path by monkeypatching the gate
        for p in blk.parameters():
            p.grad = None
        blk.conv1[1].running_mean.zero_(); blk.conv1[1].running_var.fill_(1)
        blk.conv2[1].running_mean.zero_(); blk.conv2[1].running_var.fill_(1)
        if blk.down is not None:
            blk.down[1].running_mean.zero_(); blk.down[1].running_var.fill_(1)
        blk._can_fuse = lambda _x: False
        out_u = blk(x2)
        loss_u = out_u.float().square().mean()
        loss_u.backward()

        assert torch.allclose(out_f.float(), out_u.float(), atol=3e-2, rtol=3e-2)
        assert torch.allclose(xf.float(), x2.grad.float(), atol=3e-2, rtol=3e-2)
        for a, p in zip(gf, blk.parameters()):
            assert torch.allclose(a.float(), p.grad.float(), atol=5e-2, rtol=5e-2)
        assert torch.allclose(
            rm_f, blk.conv1[1].running_mean, atol=1e-3, rtol=1e-3
        )


@pytest.mark.gpu
def test_fused_bottleneck_matches_unfused():
    from pytorch_ddp_template_amd.models.resnet import Bottleneck

    torch.manual_seed(0)
    for in_ch, ch, stride in [(64, 16, 1), (64, 32, 2)]:
        blk = Bottleneck(in_ch, ch, stride).to(torch.bfloat16).cuda()
        blk.train()
        x = torch.randn(8, 8, 8, in_ch, dtype=torch.bfloat16, device="cuda")
        x1 = x.clone().requires_grad_(True)
        x2 = x.clone().requires_grad_(True)
        out_f = blk(x1)
        out_f.float().square().mean().backward()
        gf = [p.grad.clone() for p in blk.parameters()]
        xf = x1.grad.clone()
        for p in blk.parameters():
            p.grad = None
        for seq in [blk.conv1, blk.conv2, blk.conv3] + ([blk.down] if blk.down is not None else []):
            seq[1].running_mean.zero_(); seq[1].running_var.fill_(1)
        blk._can_fuse = lambda _x: False
        out_u = blk(x2)
        out_u.float().square().mean().backward()
        assert torch.allclose(out_f.float(), out_u.float(), atol=3e-2, rtol=3e-2)
        assert torch.allclose(xf.float(), x2.grad.float(), atol=3e-2, rtol=3e-2)
        for a, p in zip(gf, blk.parameters()):
            assert torch.allclose(a.float(), p.grad.float(), atol=5e-2, rtol=5e-2)
