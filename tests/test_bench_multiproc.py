"""bench.py under torch.distributed.run with world=2 (gloo, CPU) —
validates the exact launch path the benchmark driver uses for N>1."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
@pytest.mark.parametrize("model,extra", [
    ("resnet50", ["--image-size", "64", "--num-classes", "10"]),
    ("deepfm", ["--table-rows", "10000"]),
])
def test_bench_two_ranks_gloo(model, extra):
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", "--nproc-per-node", "2",
        "--master-addr", "127.0.0.1", "--master-port", str(port),
        "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
        "--model", model, "--batch-size", "4",
    ] + extra
    r = subprocess.run(
        cmd, env=dict(os.environ, PYTHONPATH=REPO), cwd=REPO,
        capture_output=True, text=True, timeout=280,
    )
    assert r.returncode == 0, (r.stdout + r.stderr)[-4000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["n_gpus"] == 2
    assert rec["value"] > 0
    assert rec["config"]["global_batch"] == 8
