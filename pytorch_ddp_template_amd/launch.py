"""torchrun-style multi-process launcher (one OS process per GPU).

Replaces the reference's deprecated ``python -m torch.distributed.launch``
(reference run.sh:11) with a native launcher that uses the modern env
contract only (RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT —
no ``--local_rank`` argv injection needed, though the trainee still accepts
the flag), propagates the first child failure by terminating the remaining
ranks, and reaps every child (the failure-detection behavior SURVEY.md §5.3
calls out as missing from the reference).

Usage:
    python -m pytorch_ddp_template_amd.launch \
        --nproc_per_node 8 [--nnodes 1 --node_rank 0] \
        [--master_addr 127.0.0.1 --master_port 9315] \
        script.py [script args...]
"""

from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time


def build_parser():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--nproc_per_node", type=int, default=1)
    p.add_argument("--nnodes", type=int, default=1)
    p.add_argument("--node_rank", type=int, default=0)
    p.add_argument("--master_addr", type=str, default="127.0.0.1")
    p.add_argument("--master_port", type=int, default=9315)
    p.add_argument("--module", "-m", action="store_true",
                   help="treat 'script' as a python module name")
    p.add_argument("script", type=str)
    p.add_argument("script_args", nargs=argparse.REMAINDER)
    return p


def launch(args) -> int:
    world_size = args.nproc_per_node * args.nnodes
    procs: list[subprocess.Popen] = []
    base_env = dict(os.environ)
    base_env["MASTER_ADDR"] = args.master_addr
    base_env["MASTER_PORT"] = str(args.master_port)
    base_env["WORLD_SIZE"] = str(world_size)

    for local_rank in range(args.nproc_per_node):
        rank = args.node_rank * args.nproc_per_node + local_rank
        env = dict(base_env)
        env["RANK"] = str(rank)
        env["LOCAL_RANK"] = str(local_rank)
        cmd = [sys.executable]
        if args.module:
            cmd += ["-m", args.script]
        else:
            cmd += [args.script]
        cmd += ["--local_rank", str(local_rank)]
        cmd += args.script_args
        procs.append(subprocess.Popen(cmd, env=env))

    failed_rc = 0
    try:
        while procs:
            alive = []
            for p in procs:
                rc = p.poll()
                if rc is None:
                    alive.append(p)
                elif rc != 0 and failed_rc == 0:
                    failed_rc = rc
                    # first failure: take down the rest of the job rather
                    # than letting surviving ranks hang in a collective
                    for q in procs:
                        if q.poll() is None:
                            q.terminate()
            procs = alive
            if procs:
                time.sleep(0.2)
    except KeyboardInterrupt:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGINT)
        for p in procs:
            p.wait()
        raise
    finally:
        deadline = time.time() + 10
        for p in procs:
            timeout = max(0.1, deadline - time.time())
            try:
                p.wait(timeout=timeout)
            except subprocess.TimeoutExpired:
                p.kill()
                p.wait()
    return failed_rc


def main(argv=None):
    args = build_parser().parse_args(argv)
    sys.exit(launch(args))


if __name__ == "__main__":
    main()
