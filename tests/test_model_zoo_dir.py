"""--model_zoo directory resolution (reference: common/model_utils.py:27-60
— model_def resolves inside the user's model-zoo directory; the zoo dir
travels CLI -> master -> worker command line).
"""

import os
import subprocess
import sys
import tempfile
import textwrap

import pytest
import torch

from elasticdl_amd.common.args import parse_master_args
from elasticdl_amd.master.master import Master
from elasticdl_amd.utils.model_utils import get_model_spec, load_module

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

ZOO_MODULE = textwrap.dedent(
    """
    import torch
    import torch.nn as nn

    def custom_model(hidden=16, **kw):
        return nn.Sequential(nn.Linear(8, hidden), nn.ReLU(),
                             nn.Linear(hidden, 2))

    def loss(outputs, labels):
        return nn.functional.cross_entropy(outputs, labels.long())

    def optimizer(model=None):
        return ("sgd", "learning_rate=0.1")

    def eval_metrics_fn():
        return {"accuracy":
                lambda out, lab: (out.argmax(1) == lab).float().mean()}

    def feed(batch, device, dtype=None):
        x, y = batch
        x = x.to(device)
        if dtype is not None:
            x = x.to(dtype)
        return x, y.to(device)

    def synthetic_batch(batch_size=32, seed=None):
        g = torch.Generator().manual_seed(seed) if seed is not None else None
        x = torch.randn(batch_size, 8, generator=g)
        y = torch.randint(0, 2, (batch_size,), generator=g)
        return x, y
    """
)


@pytest.fixture
def zoo(tmp_path):
    (tmp_path / "linear.py").write_text(ZOO_MODULE)
    pkg = tmp_path / "vision"
    pkg.mkdir()
    (pkg / "tiny.py").write_text(ZOO_MODULE)
    return str(tmp_path)


def test_load_from_zoo_dir(zoo):
    spec = get_model_spec("linear", model_zoo=zoo)
    model = spec.build_model()
    assert model(torch.randn(4, 8)).shape == (4, 2)
    # dotted path maps to a nested file inside the zoo
    mod = load_module("vision.tiny", model_zoo=zoo)
    assert mod.custom_model()(torch.randn(2, 8)).shape == (2, 2)
    # builtin short names still win over the zoo dir
    assert get_model_spec("mnist", model_zoo=zoo).module.__name__.endswith(
        "models.mnist")
    # model_params flow through to custom_model(**params)
    spec = get_model_spec("linear", {"hidden": 4}, model_zoo=zoo)
    assert spec.build_model()[0].out_features == 4


def test_worker_command_carries_model_zoo(zoo):
    args = parse_master_args([
        "--pod_manager", "none",
        "--model_zoo", zoo,
        "--model_def", "linear",
        "--num_workers", "0",
        "--training_data", "synthetic:32",
    ])
    m = Master(args)
    cmd = m.worker_command(0)
    assert cmd[cmd.index("--model_zoo") + 1] == zoo
    assert cmd[cmd.index("--model_def") + 1] == "linear"
    m.request_stop()


@pytest.mark.timeout(300)
def test_local_job_with_custom_zoo_dir(zoo, tmp_path):
    """Full subprocess job: the worker process resolves the model from
    the zoo directory it was handed on its command line."""
    export = str(tmp_path / "model.pt")
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", zoo,
        "--model_def", "linear",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1",
        "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:96",
        "--device", "cpu",
        "--output", export,
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    assert os.path.exists(export)


@pytest.mark.timeout(300)
def test_predict_only_job_subprocess(zoo, tmp_path):
    """Predict-only job through real processes: --prediction_data is
    sharded by the master, forwarded to the worker command line, and the
    worker reads those shards through its reader chain (this used to
    leave the worker with no reader at all)."""
    outfile = str(tmp_path / "preds.txt")
    (tmp_path / "zoo2").mkdir()
    mod = ZOO_MODULE + textwrap.dedent(
        """
        def process_predictions(outputs):
            with open(%r, "a") as f:
                f.write("%%d\\n" %% outputs.shape[0])
        """ % outfile
    )
    (tmp_path / "zoo2" / "linear.py").write_text(mod)
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", str(tmp_path / "zoo2"),
        "--model_def", "linear",
        "--job_type", "predict",
        "--num_workers", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--prediction_data", "synthetic:64",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    with open(outfile) as f:
        assert sum(int(x) for x in f.read().split()) == 64


@pytest.mark.timeout(300)
def test_evaluate_only_job_subprocess(zoo, tmp_path):
    """Evaluate-only job through real processes: checkpoint restore +
    eval tasks from --validation_data, no training."""
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", zoo,
        "--model_def", "linear",
        "--job_type", "evaluate",
        "--num_workers", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--validation_data", "synthetic:64",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    assert "accuracy" in (r.stdout + r.stderr)


@pytest.mark.timeout(300)
def test_checkpoint_restore_full_stack(tmp_path):
    """--checkpoint_dir_for_init through real processes: a hand-written
    checkpoint with a perfect separator is restored by the PS daemon,
    the worker adopts the PS params (not its own random init), and an
    evaluate-only job reports accuracy 1.0."""
    from elasticdl_amd.utils.save_utils import CheckpointSaver

    zoo2 = tmp_path / "zoo"
    zoo2.mkdir()
    (zoo2 / "sep.py").write_text(textwrap.dedent(
        """
        import torch
        import torch.nn as nn

        def custom_model(**kw):
            return nn.Sequential(nn.Linear(8, 2))

        def loss(outputs, labels):
            return nn.functional.cross_entropy(outputs, labels.long())

        def optimizer(model=None):
            return ("sgd", "learning_rate=0.01")

        def eval_metrics_fn():
            return {"accuracy":
                    lambda out, lab: (out.argmax(1) == lab).float().mean()}

        def feed(batch, device, dtype=None):
            x, y = batch
            x = x.to(device)
            if dtype is not None:
                x = x.to(dtype)
            return x, y.to(device)

        def synthetic_batch(batch_size=32, seed=None):
            g = torch.Generator().manual_seed(seed) \\
                if seed is not None else None
            x = torch.randn(batch_size, 8, generator=g)
            y = (x[:, 0] > 0).long()
            return x, y
        """
    ))
    # perfect separator on y = (x0 > 0): logits = [-10*x0, +10*x0]
    w = torch.zeros(2, 8)
    w[0, 0], w[1, 0] = -10.0, 10.0
    ckpt = tmp_path / "ckpt"
    CheckpointSaver(str(ckpt)).save_shard(
        version=7,
        state={"version": 7,
               "dense": {"0.weight": w, "0.bias": torch.zeros(2)},
               "embedding_tables": {}, "embedding_infos": []},
        shard_id=0, num_shards=1,
    )
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", str(zoo2),
        "--model_def", "sep",
        "--job_type", "evaluate",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--validation_data", "synthetic:64",
        "--checkpoint_dir_for_init", str(ckpt),
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    out = r.stdout + r.stderr
    import re
    m = re.search(r"accuracy[^0-9]*([01]\.\d+)", out)
    assert m, out[-2000:]
    assert float(m.group(1)) == 1.0, m.group(1)


@pytest.mark.timeout(300)
def test_train_with_interleaved_eval_subprocess(zoo, tmp_path):
    """Training job with --evaluation_steps: version-triggered eval tasks
    interleave with training through real processes and the master logs
    aggregated metrics."""
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", zoo,
        "--model_def", "linear",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--num_epochs", "2",
        "--shuffle", "true",
        "--training_data", "synthetic:128",
        "--validation_data", "synthetic:32",
        "--evaluation_steps", "3",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    assert "accuracy" in (r.stdout + r.stderr)


@pytest.mark.timeout(300)
def test_checkpoint_write_then_resume_subprocess(zoo, tmp_path):
    """Train to --max_step with periodic PS checkpoints, then resume
    from them: PS params restore, master fast-forwards completed steps,
    and the resumed job trains only the remainder."""
    ckpt = str(tmp_path / "ckpt")
    base = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", zoo,
        "--model_def", "linear",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "1",
        "--num_epochs", "8",
        "--training_data", "synthetic:128",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(base + ["--max_step", "4",
                               "--checkpoint_dir", ckpt,
                               "--checkpoint_steps", "2"],
                       env=env, cwd=REPO, capture_output=True, text=True,
                       timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    versions = sorted(d for d in os.listdir(ckpt) if d.startswith("version-"))
    assert versions, "no checkpoints written"

    r = subprocess.run(base + ["--max_step", "6",
                               "--checkpoint_dir_for_init", ckpt],
                       env=env, cwd=REPO, capture_output=True, text=True,
                       timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]


@pytest.mark.timeout(300)
def test_two_ps_two_workers_subprocess(tmp_path):
    """Sharded PS through real processes: dense params split by name
    hash and embedding rows by id-mod across 2 PS daemons, 2 workers
    pushing concurrently; the job trains and exports."""
    export = str(tmp_path / "model.pt")
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "deepfm",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "2", "--num_ps_pods", "2",
        "--minibatch_size", "32",
        "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:256",
        "--device", "cpu",
        "--output", export,
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    assert os.path.exists(export)


@pytest.mark.timeout(300)
def test_recordio_training_job_subprocess(tmp_path):
    """Train mnist from RecordIO files on disk through real processes:
    generator -> directory of chunked files -> factory sniffing ->
    shard creation from the index -> workers decode records."""
    from elasticdl_amd.data.recordio_gen import gen_mnist_recordio

    data_dir = str(tmp_path / "mnist_rio")
    files = gen_mnist_recordio(data_dir, n=192, records_per_file=64)
    assert len(files) == 3
    export = str(tmp_path / "model.pt")
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "mnist",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", data_dir,
        "--device", "cpu",
        "--output", export,
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    assert os.path.exists(export)


@pytest.mark.timeout(300)
def test_csv_training_job_subprocess(tmp_path):
    """Train iris from a CSV file on disk through real processes
    (TextReader line-range shards, zoo collate parses rows)."""
    import random

    rng = random.Random(0)
    csv_path = tmp_path / "iris.csv"
    with open(csv_path, "w") as f:
        f.write("f0,f1,f2,f3,label\n")
        for _ in range(120):
            c = rng.randint(0, 2)
            f.write(",".join(f"{rng.gauss(c, 0.3):.3f}" for _ in range(4))
                    + f",{c}\n")
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "iris",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", str(csv_path),
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]


@pytest.mark.timeout(300)
def test_broken_model_fails_job_cleanly(zoo, tmp_path):
    """A deterministically-broken zoo module must FAIL the job quickly
    (fail-fast minibatch retries -> task retries <=3 -> all workers
    failed -> master exit != 0), not hang for the full 64-retry budget."""
    (tmp_path / "bad").mkdir()
    (tmp_path / "bad" / "broken.py").write_text(
        ZOO_MODULE.replace(
            "return nn.functional.cross_entropy(outputs, labels.long())",
            "raise RuntimeError('bad loss')",
        )
    )
    import time

    t0 = time.monotonic()
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", str(tmp_path / "bad"),
        "--model_def", "broken",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:64",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    elapsed = time.monotonic() - t0
    assert r.returncode != 0
    assert elapsed < 120, f"failure took {elapsed:.0f}s"


@pytest.mark.timeout(300)
def test_worker_crash_exhausts_relaunch_budget(tmp_path):
    """A worker that dies at startup is relaunched up to
    --relaunch_on_worker_failure times; then the job fails (exit 1,
    'All workers failed')."""
    (tmp_path / "z").mkdir()
    (tmp_path / "z" / "wcrash.py").write_text(
        "import os\n"
        "if os.environ.get('EDL_WORKER_ID') is not None:\n"
        "    raise SystemExit(3)\n" + ZOO_MODULE
    )
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", str(tmp_path / "z"),
        "--model_def", "wcrash",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--training_data", "synthetic:64",
        "--relaunch_on_worker_failure", "2",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 1
    out = r.stdout + r.stderr
    assert out.count("Relaunching failed worker") == 2
    assert "All workers failed" in out


@pytest.mark.timeout(300)
def test_ps_death_fails_job_fast_local():
    """PS failure is NOT tolerated (reference pod_event_callbacks.py:
    118-150): SIGKILL the PS daemon mid-job; the master stops the job
    with exit 1 within seconds."""
    import signal
    import time

    tmp = tempfile.mkdtemp()
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "mnist",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--training_data", "synthetic:20000",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--checkpoint_dir", tmp,
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    p = subprocess.Popen(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                         cwd=REPO, stdout=subprocess.PIPE,
                         stderr=subprocess.STDOUT, text=True)
    pid_file = os.path.join(tmp, "logs", "ps-0.pid")
    deadline = time.monotonic() + 60
    while time.monotonic() < deadline and not os.path.exists(pid_file):
        time.sleep(0.2)
    time.sleep(5)  # let the job actually start training
    os.kill(int(open(pid_file).read()), signal.SIGKILL)
    t0 = time.monotonic()
    out, _ = p.communicate(timeout=60)
    assert p.returncode == 1, out[-2000:]
    assert time.monotonic() - t0 < 30
    assert "died; stopping job" in out


@pytest.mark.timeout(300)
def test_hung_worker_killed_and_job_completes(tmp_path):
    """Task-timeout watchdog end-to-end (reference task_manager.py:
    592-616 + master.py:46-49): worker 0 hangs forever in its first
    minibatch; the watchdog requeues its task and kills the pod; the
    relaunched worker finishes the job."""
    (tmp_path / "z").mkdir()
    (tmp_path / "z" / "hang.py").write_text(ZOO_MODULE + textwrap.dedent(
        """
        import os, time as _t
        _real_loss = loss

        def loss(outputs, labels):
            if os.environ.get("EDL_WORKER_ID") == "0":
                _t.sleep(3600)
            return _real_loss(outputs, labels)
        """
    ))
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", str(tmp_path / "z"),
        "--model_def", "hang",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:64",
        "--task_timeout_sec", "6",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    out = r.stdout + r.stderr
    assert r.returncode == 0, out[-3000:]
    assert "considered hung" in out


@pytest.mark.timeout(300)
def test_data_reader_params_csv_delimiter(tmp_path):
    """--data_reader_params reaches the reader factory (reference flag:
    elasticdl_client/common/args.py:501): train iris from a
    semicolon-delimited CSV."""
    import random

    rng = random.Random(0)
    csv_path = tmp_path / "iris_semi.csv"
    with open(csv_path, "w") as f:
        f.write("f0;f1;f2;f3;label\n")
        for _ in range(96):
            c = rng.randint(0, 2)
            f.write(";".join(f"{rng.gauss(c, 0.3):.3f}" for _ in range(4))
                    + f";{c}\n")
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "iris",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", str(csv_path),
        "--data_reader_params", "delimiter=';'",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]


@pytest.mark.timeout(300)
def test_job_command_sdk_worker(tmp_path):
    """--job_command (reference pod_manager.py:327-380): worker pods run
    the user's own program, which coordinates through EDL_* env +
    MasterClient/DataShardService — the SDK-job launch path."""
    marker = str(tmp_path / "done.txt")
    script = tmp_path / "sdk_worker.py"
    script.write_text(textwrap.dedent(f"""
        import os, time
        from elasticdl_amd.worker.master_client import MasterClient
        from elasticdl_amd.worker.data_shard_service import DataShardService
        from elasticdl_amd.common.task import TaskType

        mc = MasterClient(os.environ["EDL_MASTER_ADDR"],
                          int(os.environ["EDL_WORKER_ID"]))
        svc = DataShardService(mc, 16)
        n = 0
        while True:
            t = svc.fetch_task()
            if t.type == TaskType.NONE:
                break
            if t.type == TaskType.WAIT:
                time.sleep(0.2)
                continue
            n += t.shard.size
            svc.report_batch_done(t.shard.size)
        open({marker!r}, "w").write(str(n))
    """))
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_def", "mnist",
        "--num_workers", "1",
        "--num_minibatches_per_task", "2",
        "--minibatch_size", "16",
        "--training_data", "synthetic:96",
        "--job_command", f"{sys.executable} {script}",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
    assert os.path.exists(marker)
    assert open(marker).read() == "96"


@pytest.mark.timeout(300)
def test_function_name_selector_flags(zoo, tmp_path):
    """--loss/--optimizer/... select custom function names in the module
    (reference flags, elasticdl_client/common/args.py): a module whose
    loss is called my_loss trains when --loss my_loss is passed."""
    (tmp_path / "z").mkdir()
    (tmp_path / "z" / "named.py").write_text(
        ZOO_MODULE.replace("def loss(", "def my_loss(")
                  .replace("def optimizer(", "def build_opt(")
    )
    cmd = [
        sys.executable, "-m", "elasticdl_amd.master.main",
        "--model_zoo", str(tmp_path / "z"),
        "--model_def", "named",
        "--loss", "my_loss",
        "--optimizer", "build_opt",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16",
        "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:64",
        "--device", "cpu",
        "--pod_manager", "local",
    ]
    r = subprocess.run(cmd, env=dict(os.environ, PYTHONPATH=REPO),
                       cwd=REPO, capture_output=True, text=True, timeout=280)
    assert r.returncode == 0, r.stderr[-3000:]
