import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_zoo_init(tmp_path):
    from elasticdl_amd.client import api

    api.init_zoo(str(tmp_path))
    assert (tmp_path / "model.py").exists()
    assert (tmp_path / "Dockerfile").exists()
    # the template module satisfies the zoo contract
    from elasticdl_amd.utils.model_utils import get_model_spec

    spec = get_model_spec(str(tmp_path / "model.py"))
    model = spec.build_model()
    assert model is not None


def test_cli_parser():
    from elasticdl_amd.client.main import build_parser

    p = build_parser()
    args = p.parse_args([
        "train", "--model_def", "mnist", "--num_workers", "2",
        "--distribution_strategy", "AllreduceStrategy",
    ])
    assert args.command == "train"
    assert args.num_workers == 2


@pytest.mark.timeout(240)
def test_cli_train_local_end_to_end(tmp_path):
    """`elasticdl train` without an image runs the whole job locally."""
    cmd = [
        sys.executable, "-m", "elasticdl_amd.client.main", "train",
        "--model_def", "mnist",
        "--distribution_strategy", "ParameterServerStrategy",
        "--num_workers", "1", "--num_ps_pods", "1",
        "--minibatch_size", "16", "--num_minibatches_per_task", "2",
        "--training_data", "synthetic:64",
        "--device", "cpu",
        "--checkpoint_dir", str(tmp_path),
    ]
    r = subprocess.run(
        cmd, env=dict(os.environ, PYTHONPATH=REPO), cwd=REPO,
        capture_output=True, text=True, timeout=220,
    )
    assert r.returncode == 0, r.stderr[-3000:]
