"""RCCL-path tests on a single GPU (world_size=1).

The driver benches 1/2/4/8-GPU DDP at round end; these tests catch
RCCL/stream/reducer API breakage on the nccl(=RCCL) backend before that —
wrap-time broadcast, bucketed all-reduce launch from the C++ reducer's
autograd hooks, no_sync accumulation, and a full engine step.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture()
def nccl_pg():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29771")
    torch.distributed.init_process_group("nccl", rank=0, world_size=1)
    yield
    torch.distributed.destroy_process_group()


def test_distributed_model_rccl_ws1(nccl_pg):
    from pytorch_ddp_template_amd.models import resnet18
    from pytorch_ddp_template_amd.ops import CrossEntropyLoss
    from pytorch_ddp_template_amd.optim import SGD, clip_grad_norm_
    from pytorch_ddp_template_amd.parallel import DistributedModel

    torch.manual_seed(0)
    model = resnet18(num_classes=10, stem="cifar").to(torch.bfloat16).cuda()
    dm = DistributedModel(model, bucket_bytes=4 << 20)
    opt = SGD(model.parameters(), lr=0.01, momentum=0.9, master_weights=True)
    crit = CrossEntropyLoss()
    x = torch.randn(32, 32, 32, 3, dtype=torch.bfloat16, device="cuda")
    y = torch.randint(0, 10, (32,), device="cuda")
    losses = []
    for step in range(4):
        out = dm(x)
        loss = crit(out, y)
        loss.backward()
        dm.finish_gradient_sync()
        clip_grad_norm_(list(model.parameters()), 1000.0)
        opt.step()
        dm.zero_grad()
        losses.append(float(loss))
    torch.cuda.synchronize()
    assert losses[-1] < losses[0], losses
    # grads live as views into reducer buckets; after zero_grad they are zero
    for p in model.parameters():
        assert p.grad is not None
        assert float(p.grad.abs().max()) == 0.0


def test_no_sync_accumulation_rccl(nccl_pg):
    from pytorch_ddp_template_amd.models import FooModel
    from pytorch_ddp_template_amd.parallel import DistributedModel

    torch.manual_seed(1)
    m = FooModel().cuda()
    dm = DistributedModel(m)
    x = torch.randn(8, 10, device="cuda")
    y = torch.randn(8, 5, device="cuda")
    with dm.no_sync():
        ((dm(x) - y) ** 2).mean().backward()
    g1 = [p.grad.clone() for p in m.parameters()]
    ((dm(x) - y) ** 2).mean().backward()
    dm.finish_gradient_sync()
    for a, p in zip(g1, m.parameters()):
        # second backward accumulated on top of the first
        assert torch.allclose(p.grad, 2 * a, rtol=1e-4, atol=1e-5)
