"""SDK-style custom training loop through ElasticAllReduceController
(reference zoo contract: train(dataset, elastic_controller) —
model_zoo/mnist/mnist_train_tfv2.py:21-40)."""

import torch

from elasticdl_amd.collective.controller import ElasticAllReduceController
from elasticdl_amd.collective.distributed_optimizer import DistributedOptimizer
from elasticdl_amd.common import rpc
from elasticdl_amd.master.rendezvous import ElasticRendezvousServer
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager
from elasticdl_amd.worker.master_client import MasterClient


class _FakePodManager:
    def get_alive_worker_num(self):
        return 1


def test_sdk_elastic_run_single_worker():
    tm = TaskManager()  # worker-driven shards
    rdzv = ElasticRendezvousServer("127.0.0.1")
    rdzv._flip_delay_sec = 0.0
    port = rdzv.start()
    servicer = MasterServicer(tm, rendezvous_server=rdzv,
                              pod_manager=_FakePodManager())
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        mc.report_training_params(dataset_size=64, batch_size=8,
                                  num_minibatches_per_shard=2)
        model = torch.nn.Linear(4, 1)
        opt = DistributedOptimizer(model, lr=0.05)
        ctl = ElasticAllReduceController(mc, model, opt, batch_size=8)
        ctl.start()

        x = torch.randn(8, 4)
        y = torch.randn(8)

        @ctl.elastic_run
        def train_one(x, y):
            opt.zero_grad()
            loss = torch.nn.functional.mse_loss(model(x).squeeze(-1), y)
            loss.backward()
            opt.step()
            return loss

        from elasticdl_amd.common.task import TaskType

        batches = 0
        while True:
            task = ctl.data_shard_service.fetch_task()
            if task.type != TaskType.TRAINING:
                break
            # 16 records per task, batch 8 -> 2 batches
            for _ in range(task.shard.size // 8):
                train_one(x, y)
                batches += 1
        ctl.stop()
        assert batches == 8
        assert tm.finished()
        assert ctl.global_completed_batch_num == 8
        assert ctl.comm.rendezvous_id >= 1  # communicator formed via master
    finally:
        server.stop(0)
