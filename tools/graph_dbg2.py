"""Deep probe for the batch-256 gradient-garbage bug.

Eager steps only.  After each backward, prints per-parameter grad norms for
any param whose grad is non-finite or huge, counts non-finite entries in
params/grads/momentum, and recomputes a FRESH forward loss after the
optimizer step to cross-check the printed loss.
--torch-opt uses torch's clip+SGD on the bf16 params (isolates our
multi_tensor kernels from the backward kernels).
"""
import argparse
import sys

import torch

sys.path.insert(0, "/root/repo")
from pytorch_ddp_template_amd.models import build_model
from pytorch_ddp_template_amd.ops import CrossEntropyLoss
from pytorch_ddp_template_amd.optim import SGD, clip_grad_norm_


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--lr", type=float, default=0.01)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--torch-opt", action="store_true")
    ap.add_argument("--no-fuse", action="store_true",
                    help="force the composed (non-fused-block) resnet path")
    args = ap.parse_args()

    if args.no_fuse:
        from pytorch_ddp_template_amd.models import resnet as _rn
        _rn.BasicBlock._can_fuse = lambda self, x: False
        _rn.Bottleneck._can_fuse = _rn.BasicBlock._can_fuse

    dev = torch.device("cuda:0")
    torch.manual_seed(42)
    model = build_model("resnet18", 10).to(torch.bfloat16).to(dev)
    names = [n for n, _ in model.named_parameters()]
    params = [p for _, p in model.named_parameters()]
    if args.torch_opt:
        opt = torch.optim.SGD(params, lr=args.lr, momentum=0.9,
                              weight_decay=5e-5)
    else:
        opt = SGD(params, lr=args.lr, momentum=0.9, weight_decay=5e-5,
                  master_weights=True)
    crit = CrossEntropyLoss()
    g = torch.Generator(device="cpu").manual_seed(1234)
    xs = [torch.randn(args.batch, 32, 32, 3, generator=g)
          .to(torch.bfloat16).to(dev) for _ in range(4)]
    ys = [torch.randint(0, 10, (args.batch,), generator=g).to(dev)
          for _ in range(4)]

    for i in range(args.steps):
        x, y = xs[i % 4], ys[i % 4]
        out = model(x)
        loss = crit(out, y)
        loss.backward()
        torch.cuda.synchronize()
        bad = []
        for n, p in zip(names, params):
            if p.grad is None:
                continue
            gf = p.grad.float()
            nf = int((~torch.isfinite(gf)).sum())
            gn = float(gf.norm())
            if nf or gn > 100.0 or gn != gn:
                bad.append(f"{n}: gnorm={gn:.4g} nonfinite={nf}")
        print(f"step {i} loss={float(loss.detach()):.6g} badgrads={len(bad)}",
              flush=True)
        for b in bad[:12]:
            print("   ", b, flush=True)
        if args.torch_opt:
            torch.nn.utils.clip_grad_norm_(params, 1000.0)
            opt.step()
        else:
            clip_grad_norm_(params, 1000.0)
            opt.step()
        opt.zero_grad(set_to_none=False)
        torch.cuda.synchronize()
        p_bad = sum(int((~torch.isfinite(p.float())).sum()) for p in params)
        with torch.no_grad():
            fresh = float(crit(model(xs[1]), ys[1]).detach())
        print(f"   post-step: param_nonfinite={p_bad} fresh_loss={fresh:.6g}",
              flush=True)


if __name__ == "__main__":
    main()
