"""Native C++ reducer: bucket assignment + gradient semantics (world_size=1).

World-size-2 behavior is covered by tests/test_distributed_cpu.py; here a
single-rank gloo group exercises the full machinery deterministically.
"""

import os

import pytest
import torch
import torch.distributed as dist
from torch import nn

from pytorch_ddp_template_amd.parallel import DistributedModel


@pytest.fixture
def pg(free_port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(free_port)
    dist.init_process_group("gloo", rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def small_model():
    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(10, 20), nn.ReLU(), nn.Linear(20, 5))


def test_bucket_assignment_reverse_order(pg):
    m = small_model()
    dm = DistributedModel(m, first_bucket_bytes=1, bucket_bytes=1 << 20)
    buckets = dm.reducer.bucket_assignment()
    # first bucket holds the LAST-registered parameter(s)
    n_params = len([p for p in m.parameters()])
    assert buckets[0][0] == n_params - 1
    flat = [i for b in buckets for i in b]
    assert sorted(flat) == list(range(n_params))
    # tiny first cap -> first bucket is a single param
    assert len(buckets[0]) == 1


def test_bucket_caps_split(pg):
    m = nn.Sequential(*[nn.Linear(64, 64) for _ in range(4)])
    dm = DistributedModel(m, first_bucket_bytes=64 * 64 * 4,
                          bucket_bytes=64 * 64 * 4 + 64 * 4)
    # each bucket limited to ~one weight+bias
    assert dm.reducer.num_buckets() >= 4


def test_grads_match_plain_autograd(pg):
    torch.manual_seed(1)
    m = small_model()
    ref = small_model()
    ref.load_state_dict(m.state_dict())
    dm = DistributedModel(m)
    x = torch.randn(16, 10)
    y = torch.randn(16, 5)

    out = dm(x)
    loss = ((out - y) ** 2).mean()
    loss.backward()
    dm.finish_gradient_sync()

    rout = ref(x)
    rloss = ((rout - y) ** 2).mean()
    rloss.backward()

    for p, rp in zip(m.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, rp.grad, rtol=1e-6, atol=1e-7)


def test_grad_views_are_bucket_views(pg):
    m = small_model()
    dm = DistributedModel(m)
    flats = dm.reducer.bucket_flats()
    for p in m.parameters():
        assert p.grad is not None
        assert any(
            p.grad.data_ptr() >= f.data_ptr()
            and p.grad.data_ptr() < f.data_ptr() + f.numel() * f.element_size()
            for f in flats
        )


def test_no_sync_accumulates(pg):
    torch.manual_seed(2)
    m = small_model()
    dm = DistributedModel(m)
    x1, x2 = torch.randn(8, 10), torch.randn(8, 10)

    with dm.no_sync():
        dm(x1).sum().backward()
    dm(x2).sum().backward()
    dm.finish_gradient_sync()
    acc = [p.grad.clone() for p in m.parameters()]

    dm.zero_grad()
    dm(x1).sum().backward()
    dm.finish_gradient_sync()
    g1 = [p.grad.clone() for p in m.parameters()]
    dm.zero_grad()
    dm(x2).sum().backward()
    dm.finish_gradient_sync()
    g2 = [p.grad.clone() for p in m.parameters()]

    for a, b, c in zip(acc, g1, g2):
        torch.testing.assert_close(a, b + c, rtol=1e-5, atol=1e-6)


def test_zero_grads(pg):
    m = small_model()
    dm = DistributedModel(m)
    dm(torch.randn(4, 10)).sum().backward()
    dm.finish_gradient_sync()
    assert any(p.grad.abs().sum() > 0 for p in m.parameters())
    dm.zero_grad()
    for p in m.parameters():
        assert p.grad.abs().sum() == 0


def test_find_unused_parameters(pg):
    class Branchy(nn.Module):
        def __init__(self):
            super().__init__()
            self.used = nn.Linear(10, 5)
            self.unused = nn.Linear(10, 5)

        def forward(self, x):
            return self.used(x)

    m = Branchy()
    dm = DistributedModel(m, find_unused_parameters=True)
    out = dm(torch.randn(4, 10))
    out.sum().backward()
    dm.finish_gradient_sync()  # would raise/hang if unused grads blocked it
    assert m.unused.weight.grad.abs().sum() == 0
    assert m.used.weight.grad.abs().sum() > 0


def test_second_iteration_works(pg):
    m = small_model()
    dm = DistributedModel(m)
    for _ in range(3):
        dm(torch.randn(4, 10)).sum().backward()
        dm.finish_gradient_sync()
        dm.zero_grad()


def test_multi_loss_no_sync_pattern(pg):
    """Shared-trunk / multi-loss graphs: all backward passes but the last run
    under no_sync(); grads accumulate into the bucket views and the final
    pass launches one all-reduce per bucket.  (Round-2 reducer hardening —
    the round-1 reducer hard-crashed on any second backward in a window.)"""
    torch.manual_seed(3)
    trunk = nn.Linear(10, 10)
    head_a = nn.Linear(10, 5)
    head_b = nn.Linear(10, 5)

    class TwoHead(nn.Module):
        def __init__(self):
            super().__init__()
            self.trunk, self.a, self.b = trunk, head_a, head_b

        def forward(self, x):
            t = torch.relu(self.trunk(x))
            return self.a(t), self.b(t)

    m = TwoHead()
    ref = TwoHead()
    ref.load_state_dict(m.state_dict())
    dm = DistributedModel(m)
    x = torch.randn(8, 10)
    ya, yb = dm(x)
    with dm.no_sync():
        ya.sum().backward(retain_graph=True)
    yb.sum().backward()
    dm.finish_gradient_sync()
    ra, rb = ref(x)
    (ra.sum() + rb.sum()).backward()
    for p, q in zip(m.parameters(), ref.parameters()):
        torch.testing.assert_close(p.grad, q.grad)


def test_double_backward_after_launch_raises_actionable(pg):
    """A late gradient for a bucket whose all-reduce already launched is a
    loud error pointing at no_sync(), not a cryptic crash."""
    torch.manual_seed(4)
    m = small_model()
    dm = DistributedModel(m)
    x = torch.randn(4, 10)
    out = dm(x)
    out.sum().backward(retain_graph=True)
    with pytest.raises(RuntimeError, match="no_sync"):
        out.sum().backward()
    # the reducer stays usable after the failed window
    dm.zero_grad()
    dm(x).sum().backward()
    dm.finish_gradient_sync()
