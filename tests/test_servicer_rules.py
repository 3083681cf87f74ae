"""Master servicer edge rules + PS checkpoint-during-training e2e."""

import torch

from elasticdl_amd.common import rpc
from elasticdl_amd.common.task import TaskType
from elasticdl_amd.master.rendezvous import ElasticRendezvousServer
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager


class FakePodManager:
    def __init__(self, alive):
        # accept a count (ids 0..n-1) or an explicit id list
        self.ids = list(range(alive)) if isinstance(alive, int) else list(alive)

    @property
    def alive(self):
        return len(self.ids)

    @alive.setter
    def alive(self, n):
        self.ids = list(range(n))

    def get_alive_worker_num(self):
        return len(self.ids)

    def get_alive_worker_ids(self):
        return sorted(self.ids)


def make_rdzv():
    r = ElasticRendezvousServer("127.0.0.1")
    r._flip_delay_sec = 0.0
    r._port = 1
    return r


def test_allreduce_surplus_workers_get_none_last_gets_wait():
    """Reference servicer.py:111-125: when tasks are in flight elsewhere,
    only the LAST alive worker waits; others exit (shrink the world)."""
    tm = TaskManager(training_shards=[("f", 0, 10)], records_per_task=10)
    pm = FakePodManager(alive=2)
    servicer = MasterServicer(tm, rendezvous_server=make_rdzv(), pod_manager=pm)
    t = servicer.get_task({"worker_id": 0})
    assert t["type"] == TaskType.TRAINING
    # tasks in flight, 2 alive -> the surplus (non-lowest-id) worker is
    # told to exit; the designated lowest-id worker WAITS so it can
    # drain tail tasks (train-end export) later
    assert servicer.get_task({"worker_id": 1})["type"] == TaskType.NONE
    assert servicer.get_task({"worker_id": 0})["type"] == TaskType.WAIT
    pm.alive = 1
    assert servicer.get_task({"worker_id": 0})["type"] == TaskType.WAIT


def test_ps_strategy_waits_regardless_of_alive():
    tm = TaskManager(training_shards=[("f", 0, 10)], records_per_task=10)
    servicer = MasterServicer(tm, pod_manager=FakePodManager(alive=3))
    servicer.get_task({"worker_id": 0})
    assert servicer.get_task({"worker_id": 1})["type"] == TaskType.WAIT


def test_ps_checkpoint_and_resume_mid_training(tmp_path):
    from elasticdl_amd.ps.server import ParameterServer, parse_ps_args

    args = [
        "--port", "0", "--ps_id", "0", "--num_ps_pods", "1",
        "--opt_type", "sgd", "--opt_args", "learning_rate=0.1",
        "--device", "cpu",
        "--checkpoint_dir", str(tmp_path), "--checkpoint_steps", "2",
        "--keep_checkpoint_max", "2",
    ]
    ps = ParameterServer(parse_ps_args(args))
    ps.engine.push_model({"w": torch.ones(4)}, [{"name": "emb", "dim": 4}])
    ps.engine.pull_embedding_vectors("emb", torch.tensor([1, 2, 3]))
    for _ in range(5):
        ps.engine.push_gradients({"w": torch.ones(4)}, {}, version=ps.engine.version)
    # versions 2 and 4 checkpointed
    from elasticdl_amd.utils.save_utils import latest_valid_version, list_versions

    assert latest_valid_version(str(tmp_path)) == 4
    assert list_versions(str(tmp_path)) == [2, 4]

    # resume a fresh PS from the checkpoint
    ps2 = ParameterServer(parse_ps_args(args + [
        "--checkpoint_dir_for_init", str(tmp_path),
    ]))
    assert ps2.engine.version == 4
    assert torch.allclose(ps2.engine.dense["w"], torch.ones(4) - 0.1 * 4)
    rows2 = ps2.engine.pull_embedding_vectors("emb", torch.tensor([1, 2, 3]),
                                              create=False)
    rows1 = ps.engine.pull_embedding_vectors("emb", torch.tensor([1, 2, 3]),
                                             create=False)
    assert torch.equal(rows1, rows2)
