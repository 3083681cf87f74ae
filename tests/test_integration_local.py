"""In-process distributed integration tests (the reference's primary
harness pattern: real gRPC master + PS + worker in one process,
test_utils.py:330-460), plus the full subprocess local mode."""

import os
import subprocess
import sys
import tempfile

import pytest
import torch

from elasticdl_amd.common import rpc
from elasticdl_amd.common.task import TaskType
from elasticdl_amd.master.evaluation_service import EvaluationService
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager
from elasticdl_amd.ps.server import ParameterServer, parse_ps_args
from elasticdl_amd.utils.model_utils import get_model_spec
from elasticdl_amd.worker.master_client import MasterClient
from elasticdl_amd.worker.ps_client import PSClient
from elasticdl_amd.worker.ps_trainer import ParameterServerTrainer
from elasticdl_amd.worker.worker import Worker

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def start_master(spec, reader, records_per_task=32, **tm_kw):
    tm = TaskManager(
        training_shards=reader.create_shards(),
        records_per_task=records_per_task,
        **tm_kw,
    )
    ev = EvaluationService(tm, metrics_fn=spec.eval_metrics_fn)
    servicer = MasterServicer(tm, evaluation_service=ev)
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})
    return tm, ev, servicer, server


def start_ps(num_ps, opt_type="sgd", opt_args="learning_rate=0.1", **kw):
    servers = []
    addrs = []
    for i in range(num_ps):
        argv = [
            "--port", "0", "--ps_id", str(i), "--num_ps_pods", str(num_ps),
            "--opt_type", opt_type, "--opt_args", opt_args, "--device", "cpu",
        ]
        for k, v in kw.items():
            argv += [f"--{k}", str(v)]
        ps = ParameterServer(parse_ps_args(argv))
        port = ps.start()
        servers.append(ps)
        addrs.append(f"127.0.0.1:{port}")
    return servers, addrs


def test_mnist_ps_training_in_process():
    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:128")
    tm, ev, servicer, server = start_master(spec, reader)
    ps_servers, ps_addrs = start_ps(2)
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        trainer = ParameterServerTrainer(spec, PSClient(ps_addrs), device="cpu")
        worker = Worker(0, mc, trainer, data_reader=reader, spec=spec,
                        minibatch_size=16)
        worker.run()
        assert tm.finished()
        assert tm.completed_steps == 4  # 128 / 32
        assert trainer.get_model_version() > 0
        # dense params actually live on the PS shards (2 shards)
        total = sum(len(ps.engine.dense) for ps in ps_servers)
        n_params = len([p for p in trainer.model.parameters() if p.requires_grad])
        assert total == n_params
    finally:
        server.stop(0)
        for ps in ps_servers:
            ps.server.stop(0)


def test_wide_deep_ps_training_with_embeddings():
    spec = get_model_spec("wide_deep",)
    from elasticdl_amd.data.reader import SyntheticReader
    from elasticdl_amd.models import wide_deep

    def sample(i):
        ids, labels = wide_deep.synthetic_batch(1, num_features=13, vocab=1000,
                                                seed=i)
        return ids[0], labels[0]

    reader = SyntheticReader(96, sample)
    tm, ev, servicer, server = start_master(spec, reader, records_per_task=48)
    ps_servers, ps_addrs = start_ps(2, opt_type="adam",
                                    opt_args="learning_rate=0.001")
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        trainer = ParameterServerTrainer(spec, PSClient(ps_addrs), device="cpu")
        worker = Worker(0, mc, trainer, data_reader=reader, spec=spec,
                        minibatch_size=16)
        worker.run()
        assert tm.finished()
        # embedding rows created on both shards
        rows = [
            ps.engine.tables["deep_embedding"].num_rows for ps in ps_servers
        ]
        assert all(r > 0 for r in rows), rows
    finally:
        server.stop(0)
        for ps in ps_servers:
            ps.server.stop(0)


def test_two_workers_share_tasks():
    import threading

    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:256")
    tm, ev, servicer, server = start_master(spec, reader)
    ps_servers, ps_addrs = start_ps(1)
    try:
        def run_worker(wid):
            mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=wid)
            trainer = ParameterServerTrainer(spec, PSClient(ps_addrs),
                                             device="cpu")
            Worker(wid, mc, trainer, data_reader=reader, spec=spec,
                   minibatch_size=16).run()

        threads = [threading.Thread(target=run_worker, args=(i,)) for i in range(2)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(180)
        assert tm.finished()
        assert tm.completed_steps == 8
    finally:
        server.stop(0)
        for ps in ps_servers:
            ps.server.stop(0)


def test_sync_sgd_two_workers_grads_to_wait():
    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:64")
    tm, ev, servicer, server = start_master(spec, reader)
    ps_servers, ps_addrs = start_ps(1, use_async="false", grads_to_wait=2,
                                    sync_version_tolerance=2)
    try:
        import threading

        def run_worker(wid):
            mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=wid)
            trainer = ParameterServerTrainer(spec, PSClient(ps_addrs),
                                             device="cpu")
            Worker(wid, mc, trainer, data_reader=reader, spec=spec,
                   minibatch_size=16).run()

        threads = [threading.Thread(target=run_worker, args=(i,)) for i in range(2)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(180)
        assert tm.finished()
        # sync PS: version == pushes / grads_to_wait (4 tasks x 2... but
        # workers race; just require it advanced and is consistent)
        assert ps_servers[0].engine.version >= 1
    finally:
        server.stop(0)
        for ps in ps_servers:
            ps.server.stop(0)


@pytest.mark.timeout(300)
def test_local_mode_subprocess_e2e():
    """BASELINE config 1: master + 1 worker + 1 PS as real processes."""
    with tempfile.TemporaryDirectory() as tmp:
        export = os.path.join(tmp, "model.pt")
        cmd = [
            sys.executable, "-m", "elasticdl_amd.master.main",
            "--model_def", "mnist",
            "--distribution_strategy", "ParameterServerStrategy",
            "--num_workers", "1",
            "--num_ps_pods", "1",
            "--minibatch_size", "16",
            "--num_minibatches_per_task", "2",
            "--training_data", "synthetic:96",
            "--device", "cpu",
            "--checkpoint_dir", tmp,
            "--output", export,
            "--pod_manager", "local",
        ]
        env = dict(os.environ, PYTHONPATH=REPO)
        r = subprocess.run(cmd, env=env, cwd=REPO, capture_output=True,
                           text=True, timeout=280)
        assert r.returncode == 0, r.stderr[-3000:]
        assert os.path.exists(export), "train-end export missing"
        state = torch.load(export, weights_only=True)
        assert any("net" in k for k in state)


@pytest.mark.timeout(600)
def test_ps_strategy_worker_preemption_recovers(tmp_path):
    """PS-strategy elasticity: SIGKILL a worker mid-job; the master
    recovers its task shards, relaunches a replacement (PS state lives
    on), and the job completes. (Round-1 elastic evidence covered the
    AllReduce path only.)"""
    import signal
    import subprocess
    import sys
    import time

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "mnist",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "2", "--num_ps_pods", "1",
        "--training_data", "synthetic:4096",
        "--minibatch_size", "32",
        "--num_minibatches_per_task", "2",
        "--pod_manager", "local",
        "--device", "cpu",
        "--checkpoint_dir", str(tmp_path),
    ]
    p = subprocess.Popen(
        cmd, cwd=repo, env=dict(os.environ, PYTHONPATH=repo),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
    )
    pidfile = os.path.join(str(tmp_path), "logs", "worker-0.pid")
    deadline = time.time() + 120
    wpid = None
    while time.time() < deadline and wpid is None:
        time.sleep(0.5)
        if os.path.exists(pidfile):
            with open(pidfile) as f:
                wpid = int(f.read().strip())
    assert wpid is not None, "worker-0 never started"
    time.sleep(5)  # let it take tasks mid-run
    try:
        os.kill(wpid, signal.SIGKILL)
    except ProcessLookupError:
        pass  # finished early; completion still validated below
    out, _ = p.communicate(timeout=540)
    assert p.returncode == 0, out[-4000:]


def test_single_worker_ps_actually_learns():
    """Regression: after a push the trainer must NOT claim the post-push
    model version — that satisfied the PS's pull version-gate forever,
    so a 1-worker async job computed every gradient against the initial
    weights and never learned (loss pinned at ln 2 on separable data)."""
    import types

    import torch.nn as nn

    from elasticdl_amd.data.reader import SyntheticReader
    from elasticdl_amd.utils.model_utils import ModelSpec

    torch.manual_seed(0)
    w_true = torch.randn(16)

    def sample(i):
        g = torch.Generator().manual_seed(i)
        x = torch.randn(16, generator=g)
        return x, (x @ w_true > 0).long()

    reader = SyntheticReader(6144, sample, records_per_shard=128)

    def model_fn():
        torch.manual_seed(7)
        return nn.Sequential(nn.Linear(16, 64), nn.ReLU(),
                             nn.Linear(64, 1), nn.Flatten(0))

    spec = ModelSpec(
        module=types.SimpleNamespace(),
        model_fn=model_fn,
        loss_fn=lambda o, y: torch.nn.functional
        .binary_cross_entropy_with_logits(o.float(), y.float()),
        optimizer_fn=lambda m=None: ("sgd", "learning_rate=0.05"),
    )
    tm, ev, servicer, server = start_master(spec, reader,
                                            records_per_task=128)
    ps_servers, ps_addrs = start_ps(1, opt_args="learning_rate=0.05")
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        trainer = ParameterServerTrainer(spec, PSClient(ps_addrs),
                                         device="cpu")
        losses = []
        orig = trainer.train_minibatch

        def recording(batch):
            loss, v = orig(batch)
            losses.append(float(loss))
            return loss, v

        trainer.train_minibatch = recording
        worker = Worker(0, mc, trainer, data_reader=reader, spec=spec,
                        minibatch_size=32)
        worker.run()
        first = sum(losses[:5]) / 5
        last = sum(losses[-5:]) / 5
        # pre-fix behavior: loss pinned at ~ln2 = 0.693 forever
        assert last < 0.75 * first and last < 0.55, (first, last)
    finally:
        server.stop(0)
        for ps in ps_servers:
            ps.server.stop(0)


def test_get_model_steps_local_updates():
    """train_with_local_model (reference worker.py:305-388): with
    --get_model_steps 4 the worker pulls dense params every 4th step and
    applies local-optimizer updates in between — and the 1-worker async
    job still learns (the local updates carry the progress between
    pulls)."""
    spec = get_model_spec("iris")
    import torch

    torch.manual_seed(0)

    def sample(i):
        g = torch.Generator().manual_seed(i)
        c = i % 3
        x = torch.randn(4, generator=g) * 0.25 + float(c)
        return [*x.tolist(), c]

    from elasticdl_amd.data.reader import SyntheticReader

    reader = SyntheticReader(4096, sample, name="iris-sep")
    tm, ev, servicer, server = start_master(spec, reader,
                                            records_per_task=512)
    ps_servers, ps_addrs = start_ps(1, opt_args="learning_rate=0.05")
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        client = PSClient(ps_addrs)
        pulls = {"n": 0}
        real_pull = client.pull_dense_parameters

        def counted_pull(*a, **kw):
            pulls["n"] += 1
            return real_pull(*a, **kw)

        client.pull_dense_parameters = counted_pull
        trainer = ParameterServerTrainer(spec, client, device="cpu",
                                         get_model_steps=4)
        losses = []
        real_train = trainer.train_minibatch

        def recording_train(batch):
            loss, v = real_train(batch)
            losses.append(float(loss))
            return loss, v

        trainer.train_minibatch = recording_train
        worker = Worker(0, mc, trainer, data_reader=reader, spec=spec,
                        minibatch_size=32)
        worker.run()
        assert tm.finished()
        n_steps = len(losses)
        assert n_steps == 128  # 4096 / 32
        # pulls gated to every 4th step (one extra initial _get_model)
        assert pulls["n"] <= n_steps // 4 + 2, pulls["n"]
        # it learned: tail loss well below the first-step loss
        head = sum(losses[:4]) / 4
        tail = sum(losses[-4:]) / 4
        assert tail < head * 0.6, (head, tail)
    finally:
        server.stop(0)
        for ps in ps_servers:
            ps.server.stop(0)
