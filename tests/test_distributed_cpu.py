"""World-size-2 gloo CPU tests — BASELINE.json config 1 (the plumbing rung).

Covers: gradient averaging across ranks vs the manual average, wrap-time
parameter broadcast, grad-accumulation no_sync, and a 2-rank end-to-end
train-loop parity check against a single-process run at equal global batch.
"""

import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch import nn


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(10, 32), nn.ReLU(), nn.Linear(32, 5))


# ---------------- worker fns (top-level for pickling) ----------------


def _w_grad_avg(rank, world, port, q):
    from pytorch_ddp_template_amd.parallel import DistributedModel

    _init(rank, world, port)
    m = _model(seed=rank)  # different init per rank; broadcast must fix it
    dm = DistributedModel(m)
    torch.manual_seed(100 + rank)
    x = torch.randn(8, 10)
    y = torch.randn(8, 5)
    out = dm(x)
    ((out - y) ** 2).mean().backward()
    dm.finish_gradient_sync()
    grads = [p.grad.clone() for p in m.parameters()]
    params = [p.detach().clone() for p in m.parameters()]
    q.put((rank, [g.numpy() for g in grads], [p.numpy() for p in params],
           x.numpy(), y.numpy()))
    dist.barrier()
    dist.destroy_process_group()


def _w_train_loop(rank, world, port, q):
    import numpy as np

    from pytorch_ddp_template_amd.parallel import DistributedModel

    _init(rank, world, port)
    m = _model(seed=0)
    dm = DistributedModel(m)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    losses = []
    for step in range(5):
        # deterministic global batch of 8, split 4/4 across ranks
        g = torch.Generator().manual_seed(1000 + step)
        X = torch.randn(8, 10, generator=g)
        Y = torch.randn(8, 5, generator=g)
        x = X[rank * 4:(rank + 1) * 4]
        y = Y[rank * 4:(rank + 1) * 4]
        out = dm(x)
        loss = ((out - y) ** 2).mean()
        loss.backward()
        dm.finish_gradient_sync()
        opt.step()
        dm.zero_grad()
        # global loss for comparison = mean over both shards
        lt = loss.detach().clone()
        dist.all_reduce(lt)
        losses.append(float(lt) / world)
    if rank == 0:
        q.put((losses, [p.detach().numpy().copy() for p in m.parameters()]))
    dist.barrier()
    dist.destroy_process_group()


def _spawn(fn, world, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    ps = []
    for r in range(world):
        p = ctx.Process(target=fn, args=(r, world, port, q))
        p.start()
        ps.append(p)
    results = []
    for _ in range(world if fn is _w_grad_avg else 1):
        results.append(q.get())
    for p in ps:
        p.join(timeout=120)
        assert p.exitcode == 0
    return results


def test_grad_averaging_and_broadcast(free_port):
    import numpy as np

    results = _spawn(_w_grad_avg, 2, free_port)
    results.sort(key=lambda r: r[0])
    (r0, g0, p0, x0, y0), (r1, g1, p1, x1, y1) = results
    # params identical after wrap-time broadcast
    for a, b in zip(p0, p1):
        np.testing.assert_allclose(a, b, rtol=0, atol=0)
    # reduced grads identical on both ranks
    for a, b in zip(g0, g1):
        np.testing.assert_allclose(a, b, rtol=1e-5, atol=1e-6)
    # and equal to the manual average of per-shard grads
    m = _model(seed=0)
    ref = _model(seed=0)
    ref.load_state_dict(
        {k: torch.tensor(v) for (k, _), v in zip(m.state_dict().items(), p0)}
    )
    grads_manual = []
    for xx, yy in ((x0, y0), (x1, y1)):
        ref.zero_grad()
        out = ref(torch.tensor(xx))
        ((out - torch.tensor(yy)) ** 2).mean().backward()
        grads_manual.append([p.grad.clone() for p in ref.parameters()])
    for i, a in enumerate(g0):
        avg = (grads_manual[0][i] + grads_manual[1][i]) / 2
        np.testing.assert_allclose(a, avg.numpy(), rtol=1e-4, atol=1e-5)


def test_two_rank_loop_matches_single_process(free_port):
    import numpy as np

    (losses, params), = _spawn(_w_train_loop, 2, free_port)
    # single-process reference at the same global batch
    m = _model(seed=0)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    ref_losses = []
    for step in range(5):
        g = torch.Generator().manual_seed(1000 + step)
        X = torch.randn(8, 10, generator=g)
        Y = torch.randn(8, 5, generator=g)
        opt.zero_grad()
        # same global-batch gradient: mean of shard losses
        l0 = ((m(X[:4]) - Y[:4]) ** 2).mean()
        l1 = ((m(X[4:]) - Y[4:]) ** 2).mean()
        loss = (l0 + l1) / 2
        loss.backward()
        opt.step()
        ref_losses.append(float(loss))
    np.testing.assert_allclose(losses, ref_losses, rtol=1e-5, atol=1e-6)
    for p, rp in zip(params, m.parameters()):
        np.testing.assert_allclose(p, rp.detach().numpy(), rtol=1e-5, atol=1e-6)
