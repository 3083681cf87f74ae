#!/usr/bin/env python3
"""Elastic recovery benchmark (BASELINE config 5).

Runs an AllReduce job locally (master + N worker processes), preempts K
workers mid-run (SIGKILL), lets them rejoin via relaunch, and reports the
throughput timeline + seconds-to-recover. The reference only claims this
qualitatively (README.md:53-78); here it is measured.

    python scripts/bench_elastic.py --workers 4 --preempt 2 --records 4096
"""

import argparse
import json
import os
import re
import signal
import subprocess
import sys
import tempfile
import threading
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", type=int, default=4)
    ap.add_argument("--preempt", type=int, default=2)
    ap.add_argument("--records", type=int, default=4096)
    ap.add_argument("--minibatch", type=int, default=32)
    ap.add_argument("--model", default="mnist")
    ap.add_argument("--preempt-after-sec", type=float, default=20.0)
    ap.add_argument("--rejoin", action="store_true", default=True)
    args = ap.parse_args()

    tmp = os.environ.get("EDL_ELASTIC_TMP") or tempfile.mkdtemp(
        prefix="edl-elastic-"
    )
    os.makedirs(tmp, exist_ok=True)
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", args.model,
        "--distribution_strategy", "AllreduceStrategy",
        "--num_workers", str(args.workers),
        "--minibatch_size", str(args.minibatch),
        "--num_minibatches_per_task", "2",
        "--training_data", f"synthetic:{args.records}",
        "--device", "auto",
        "--checkpoint_dir", tmp,
        "--pod_manager", "local",
        "--log_loss_steps", "1",
    ]
    env = dict(os.environ, PYTHONPATH=REPO, EDL_PG_TIMEOUT_SEC="20")
    t_start = time.time()
    master = subprocess.Popen(cmd, env=env, cwd=REPO,
                              stdout=subprocess.PIPE,
                              stderr=subprocess.STDOUT, text=True)

    # scrape master stdout for pod events (exit / relaunch timeline)
    events = []
    log_lines = []

    def reader():
        for line in master.stdout:
            log_lines.append(line)
            now = time.time() - t_start
            if "exited rc=" in line or "Relaunching" in line:
                events.append((round(now, 1), line.strip().split("] ")[-1]))

    rt = threading.Thread(target=reader, daemon=True)
    rt.start()

    time.sleep(args.preempt_after_sec)
    killed_at = time.time() - t_start
    killed = 0
    for wid in range(args.preempt):
        pidfile = os.path.join(tmp, "logs", f"worker-{wid}.pid")
        if os.path.exists(pidfile):
            with open(pidfile) as f:
                pid = int(f.read().strip())
            try:
                os.kill(pid, signal.SIGKILL)
                killed += 1
            except ProcessLookupError:
                pass
    print(f"[elastic-bench] killed {killed} workers at t={killed_at:.1f}s",
          flush=True)

    rc = master.wait(timeout=1800)
    total = time.time() - t_start
    rt.join(5)

    # recovery time: first "Relaunching"/rendezvous re-formation after kill
    recover_evt = next(
        (t for t, e in events if t > killed_at and "Relaunch" in e), None
    )
    print(json.dumps({
        "metric": "elastic_recovery",
        "exit_code": rc,
        "workers": args.workers,
        "preempted": killed,
        "killed_at_sec": round(killed_at, 1),
        "relaunch_at_sec": recover_evt,
        "sec_to_recover": None if recover_evt is None
        else round(recover_evt - killed_at, 1),
        "total_sec": round(total, 1),
        "events": events[:20],
    }), flush=True)
    return rc


if __name__ == "__main__":
    sys.exit(main())
