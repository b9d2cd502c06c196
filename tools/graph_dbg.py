"""Bisect the hipGraph-captured bench step divergence at small batch.

Modes (--capture):
  none    - fully eager control
  fwdbwd  - capture forward+loss+backward only; clip/opt/zero eager
  opt     - eager fwd/bwd; capture clip+opt+zero
  full    - capture everything (the bench default path)
Flags: --no-clip drops clip_grad_norm_ from the step; --torch-sgd uses
torch.optim.SGD (fp32 model) instead of the fused kernel.
Prints per-step loss, pre-clip grad norm, and a param norm probe.
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
    ap.add_argument("--capture", choices=["none", "fwdbwd", "opt", "full"],
                    default="full")
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--lr", type=float, default=0.01)
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--no-clip", action="store_true")
    ap.add_argument("--torch-sgd", action="store_true")
    args = ap.parse_args()

    dev = torch.device("cuda:0")
    torch.manual_seed(42)
    dtype = torch.float32 if args.torch_sgd else torch.bfloat16
    model = build_model("resnet18", 10).to(dtype).to(dev)
    if args.torch_sgd:
        opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                              weight_decay=5e-5)
    else:
        opt = SGD(model.parameters(), lr=args.lr, momentum=0.9,
                  weight_decay=5e-5, master_weights=True)
    crit = CrossEntropyLoss()
    g = torch.Generator(device="cpu").manual_seed(1234)
    xs = [torch.randn(args.batch, 32, 32, 3, generator=g).to(dtype).to(dev)
          for _ in range(4)]
    ys = [torch.randint(0, 10, (args.batch,), generator=g).to(dev)
          for _ in range(4)]
    x_st, y_st = xs[0].clone(), ys[0].clone()
    lr_dev = torch.full((), args.lr, dtype=torch.float32, device=dev)
    probe = next(p for p in model.parameters() if p.numel() > 10000)

    gnorm_st = torch.zeros((), dtype=torch.float32, device=dev)

    def fwd_bwd():
        out = model(x_st)
        loss = crit(out, y_st)
        loss.backward()
        return loss

    def clip_opt_zero():
        grads = [p.grad for p in model.parameters() if p.grad is not None]
        gnorm_st.copy_(
            torch.sqrt(sum((gg.float() ** 2).sum() for gg in grads)))
        if not args.no_clip:
            clip_grad_norm_(list(model.parameters()), 1000.0)
        if args.torch_sgd:
            opt.step()
        else:
            opt.step(lr_tensor=lr_dev)
        model.zero_grad(set_to_none=False)

    def full_step():
        loss = fwd_bwd()
        clip_opt_zero()
        return loss

    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for i in range(3):
            l = full_step()
            torch.cuda.synchronize()
            print(f"warmup {i} loss={float(l.detach()):.6g} "
                  f"gnorm={float(gnorm_st):.6g} "
                  f"pnorm={float(probe.float().norm()):.6g}", flush=True)
            del l
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()

    loss_st = None
    graph = None
    if args.capture != "none":
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, stream=side):
            if args.capture == "fwdbwd":
                loss_st = fwd_bwd()
            elif args.capture == "opt":
                clip_opt_zero()
            else:
                loss_st = full_step()

    for i in range(args.steps):
        x_st.copy_(xs[i % 4])
        y_st.copy_(ys[i % 4])
        if args.capture == "none":
            loss_st = full_step()
        elif args.capture == "fwdbwd":
            graph.replay()
            clip_opt_zero()
        elif args.capture == "opt":
            loss_st = fwd_bwd()
            graph.replay()
        else:
            graph.replay()
        torch.cuda.synchronize()
        print(f"step {i} loss={float(loss_st.detach()):.6g} "
              f"gnorm={float(gnorm_st):.6g} "
              f"pnorm={float(probe.float().norm()):.6g}", flush=True)


if __name__ == "__main__":
    main()
