"""Evaluate-only and predict-only job flows + RecordIndexService."""

import torch

from elasticdl_amd.common import rpc
from elasticdl_amd.common.task import TaskType
from elasticdl_amd.master.evaluation_service import EvaluationService
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager
from elasticdl_amd.utils.model_utils import get_model_spec
from elasticdl_amd.worker.master_client import MasterClient
from elasticdl_amd.worker.trainer import LocalTrainer
from elasticdl_amd.worker.worker import Worker


def run_job(tm, spec, reader, eval_reader=None):
    ev = EvaluationService(tm, metrics_fn=spec.eval_metrics_fn)
    servicer = MasterServicer(tm, evaluation_service=ev)
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        trainer = LocalTrainer(spec, device="cpu")
        Worker(0, mc, trainer, data_reader=reader,
               eval_data_reader=eval_reader, spec=spec,
               minibatch_size=16).run()
        return ev
    finally:
        server.stop(0)


def test_evaluate_only_job():
    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:64")
    tm = TaskManager(evaluation_shards=[("mnist-synthetic", 0, 64)],
                     records_per_task=32)
    n = tm.create_evaluation_tasks(model_version=0)
    assert n == 2
    ev = run_job(tm, spec, reader, eval_reader=reader)
    assert tm.finished()
    assert "accuracy" in ev.latest_result


def test_predict_only_job():
    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:64")
    collected = []
    spec.module.process_predictions = collected.append  # zoo output hook
    try:
        tm = TaskManager(prediction_shards=[("mnist-synthetic", 0, 64)],
                         records_per_task=32)
        assert tm.create_prediction_tasks() == 2
        run_job(tm, spec, reader, eval_reader=reader)
        assert tm.finished()
        assert sum(t.shape[0] for t in collected) == 64
    finally:
        del spec.module.process_predictions


def test_record_index_service():
    from elasticdl_amd.worker.data_shard_service import RecordIndexService

    tm = TaskManager(training_shards=[("s", 0, 40)], records_per_task=20)
    servicer = MasterServicer(tm)
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        svc = RecordIndexService(mc, batch_size=10).start()
        seen = []
        while len(seen) < 40:
            idx = svc.next_index(timeout=30)
            assert idx is not None
            seen.append(idx)
            if len(seen) % 10 == 0:
                svc.report_batch_done(10)
        assert sorted(seen) == list(range(40))
        svc.stop()
        # tasks completed through report_batch_done
        import time

        deadline = time.time() + 10
        while not tm.finished() and time.time() < deadline:
            time.sleep(0.1)
        assert tm.finished()
    finally:
        server.stop(0)
