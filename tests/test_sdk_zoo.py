"""SDK-style zoo module end-to-end: user loop + elastic controller."""

import torch

from elasticdl_amd.common import rpc
from elasticdl_amd.common.task import TaskType
from elasticdl_amd.master.rendezvous import ElasticRendezvousServer
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager
from elasticdl_amd.models import mnist_sdk_train
from elasticdl_amd.worker.master_client import MasterClient


class _OnePod:
    def get_alive_worker_num(self):
        return 1


def test_sdk_zoo_trains():
    reader = mnist_sdk_train.custom_data_reader("synthetic:64")
    tm = TaskManager(training_shards=reader.create_shards(),
                     records_per_task=16)
    rdzv = ElasticRendezvousServer("127.0.0.1")
    rdzv._flip_delay_sec = 0.0
    rdzv.start()
    servicer = MasterServicer(tm, rendezvous_server=rdzv,
                              pod_manager=_OnePod())
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        model, opt, controller, device = (
            mnist_sdk_train.create_model_and_optimizer(mc, batch_size=16)
        )
        controller.start()

        def stream():
            from torch.utils.data import default_collate

            while True:
                task = controller.data_shard_service.fetch_task()
                if task.type != TaskType.TRAINING:
                    return
                records = [
                    reader.sample_fn(i)
                    for i in range(task.shard.start, task.shard.end)
                ]
                for lo in range(0, len(records), 16):
                    yield default_collate(records[lo:lo + 16])

        losses = mnist_sdk_train.train(stream(), controller, model, opt,
                                       device=device)
        controller.stop()
        assert len(losses) == 4
        assert all(l == l for l in losses)  # finite
        assert tm.finished()
    finally:
        server.stop(0)
