"""Launcher: env contract + failure propagation."""

import subprocess
import sys
import textwrap


def run_launcher(tmp_path, script_body, nproc=2, port=29511):
    script = tmp_path / "child.py"
    script.write_text(textwrap.dedent(script_body))
    return subprocess.run(
        [
            sys.executable,
            "-m",
            "pytorch_ddp_template_amd.launch",
            "--nproc_per_node",
            str(nproc),
            "--master_port",
            str(port),
            str(script),
        ],
        capture_output=True,
        text=True,
        timeout=120,
    )


def test_env_contract(tmp_path, free_port):
    r = run_launcher(
        tmp_path,
        """
        import os, sys
        print("ENV", os.environ["RANK"], os.environ["LOCAL_RANK"],
              os.environ["WORLD_SIZE"], os.environ["MASTER_ADDR"],
              os.environ["MASTER_PORT"], sys.argv[1:], flush=True)
        """,
        port=free_port,
    )
    assert r.returncode == 0, r.stderr
    lines = sorted(l for l in r.stdout.splitlines() if l.startswith("ENV"))
    assert len(lines) == 2
    assert "ENV 0 0 2 127.0.0.1" in lines[0]
    assert "ENV 1 1 2 127.0.0.1" in lines[1]
    # --local_rank passed for reference-compat
    assert "'--local_rank'" in lines[0]


def test_failure_propagates(tmp_path, free_port):
    r = run_launcher(
        tmp_path,
        """
        import os, sys, time
        if os.environ["RANK"] == "1":
            sys.exit(3)
        time.sleep(60)  # rank 0 hangs; launcher must kill it
        """,
        port=free_port,
    )
    assert r.returncode == 3
