"""Small-unit coverage: timing, rpc error paths, codec fuzz, client
degradation, worker WAIT continuation."""

import random
import threading
import time

import grpc
import pytest
import torch

from elasticdl_amd.common import codec, rpc
from elasticdl_amd.common.task import TaskType
from elasticdl_amd.common.timing import Timing


def test_timing_sections():
    t = Timing(enabled=True)
    t.start_record_time("a")
    time.sleep(0.01)
    t.end_record_time("a")
    report = t.report_timing(reset=True)
    assert "a:" in report
    assert t.acc == {}
    # disabled timing is a no-op
    t2 = Timing(enabled=False)
    t2.start_record_time("x")
    t2.end_record_time("x")
    assert t2.acc == {}


def test_codec_fuzz_roundtrip():
    rng = random.Random(0)

    def gen(depth=0):
        kind = rng.randrange(6 if depth < 3 else 4)
        if kind == 0:
            return rng.randint(-10**12, 10**12)
        if kind == 1:
            return rng.random()
        if kind == 2:
            return "".join(chr(rng.randrange(32, 1000)) for _ in range(8))
        if kind == 3:
            return torch.randn(rng.randrange(0, 5), rng.randrange(1, 4))
        if kind == 4:
            return [gen(depth + 1) for _ in range(rng.randrange(4))]
        return {f"k{i}": gen(depth + 1) for i in range(rng.randrange(4))}

    for _ in range(25):
        msg = {"payload": gen()}
        out = codec.decode(codec.encode(msg))

        def eq(a, b):
            if isinstance(a, torch.Tensor):
                return torch.equal(a, b)
            if isinstance(a, (list, tuple)):
                return len(a) == len(b) and all(eq(x, y) for x, y in zip(a, b))
            if isinstance(a, dict):
                return a.keys() == b.keys() and all(eq(a[k], b[k]) for k in a)
            if isinstance(a, float):
                return abs(a - b) < 1e-12
            return a == b

        assert eq(msg, out)


def test_rpc_handler_error_propagates_as_internal():
    def boom(req):
        raise ValueError("nope")

    server = rpc.start_server("127.0.0.1:0", {"S": {"boom": boom}})
    try:
        client = rpc.RpcClient(f"127.0.0.1:{server.port}")
        with pytest.raises(grpc.RpcError) as exc:
            client.call("S", "boom", {})
        assert exc.value.code() == grpc.StatusCode.INTERNAL
    finally:
        server.stop(0)


def test_master_client_degrades_when_master_gone():
    from elasticdl_amd.worker.master_client import MasterClient

    mc = MasterClient("127.0.0.1:1", worker_id=0)  # nothing listening
    task = mc.get_task()
    assert task.type == TaskType.NONE
    mc.report_task_result(1)  # must not raise
    info = mc.get_comm_rank()
    assert info["rank_id"] == -1
    assert mc.job_finished()


def test_worker_wait_then_tasks_appear():
    from elasticdl_amd.master.servicer import MasterServicer
    from elasticdl_amd.master.task_manager import TaskManager
    from elasticdl_amd.utils.model_utils import get_model_spec
    from elasticdl_amd.worker.master_client import MasterClient
    from elasticdl_amd.worker.trainer import LocalTrainer
    from elasticdl_amd.worker.worker import Worker

    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:32")
    tm = TaskManager()  # worker-driven: worker WAITs until params arrive
    servicer = MasterServicer(tm)
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        worker = Worker(0, mc, LocalTrainer(spec, "cpu"), data_reader=reader,
                        spec=spec, minibatch_size=16)
        t = threading.Thread(target=worker.run)
        t.start()
        time.sleep(1.0)  # worker should be WAITing
        assert not tm.finished()
        mc.report_training_params(dataset_size=32, batch_size=16)
        t.join(60)
        assert not t.is_alive()
        assert tm.finished()
    finally:
        server.stop(0)


def test_cli_serve_parser():
    from elasticdl_amd.client.main import build_parser

    args = build_parser().parse_args([
        "serve", "--model_def", "mnist", "--port", "9000",
    ])
    assert args.command == "serve"
    assert args.port == 9000


def test_task_manager_throughput_counters():
    from elasticdl_amd.master.task_manager import TaskManager

    tm = TaskManager(training_shards=[("f", 0, 64)], records_per_task=16)
    while True:
        t = tm.get(0)
        if t.type != TaskType.TRAINING:
            break
        tm.report(t.task_id, True, 0)
    c = tm.counts()
    assert c["completed_records"] == 64
    assert c["records_per_sec"] >= 0


def test_timing_sections_and_worker_timing_log(tmp_path):
    """EDL_TIMING observability (reference common/timing_utils.py:17-48 +
    per-task report, worker.py:374-376): sections accumulate and the
    worker logs a per-task timing line when enabled."""
    import os
    import subprocess
    import sys

    from elasticdl_amd.common.timing import Timing

    t = Timing(enabled=True)
    t.start_record_time("a")
    t.end_record_time("a")
    t.start_record_time("a")
    t.end_record_time("a")
    t.start_record_time("b")
    t.end_record_time("b")
    rep = t.report_timing(reset=True)
    assert "a:" in rep and "b:" in rep and "ms" in rep
    assert t.report_timing() == ""
    # disabled -> no-ops
    t2 = Timing(enabled=False)
    t2.start_record_time("x")
    t2.end_record_time("x")
    assert t2.report_timing() == ""

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "elasticdl_amd.master.main",
         "--model_def", "mnist", "--num_workers", "1",
         "--training_data", "synthetic:32", "--minibatch_size", "16",
         "--num_minibatches_per_task", "2", "--device", "cpu",
         "--checkpoint_dir", str(tmp_path),
         "--pod_manager", "local"],
        env=dict(os.environ, PYTHONPATH=repo, EDL_TIMING="1"),
        cwd=repo, capture_output=True, text=True, timeout=280,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    wlog = open(os.path.join(str(tmp_path), "logs", "worker-0.log")).read()
    assert "timing:" in wlog and "batch_process" in wlog
