"""Version-triggered evaluation end-to-end (reference flow SURVEY §3.5):
PS version hits evaluation_steps -> report_version to master -> eval
tasks -> worker forward -> metrics aggregated on master."""

from elasticdl_amd.common import rpc
from elasticdl_amd.master.evaluation_service import EvaluationService
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager
from elasticdl_amd.ps.server import ParameterServer, parse_ps_args
from elasticdl_amd.utils.model_utils import get_model_spec
from elasticdl_amd.worker.master_client import MasterClient
from elasticdl_amd.worker.ps_client import PSClient
from elasticdl_amd.worker.ps_trainer import ParameterServerTrainer
from elasticdl_amd.worker.worker import Worker


def test_eval_tasks_triggered_and_aggregated():
    spec = get_model_spec("mnist")
    reader = spec.data_reader_fn("synthetic:128")
    tm = TaskManager(
        training_shards=reader.create_shards(),
        evaluation_shards=[("mnist-synthetic", 0, 32)],
        records_per_task=32,
    )
    ev = EvaluationService(tm, evaluation_steps=2, metrics_fn=spec.eval_metrics_fn)
    servicer = MasterServicer(tm, evaluation_service=ev)
    server = rpc.start_server("127.0.0.1:0", {"Master": servicer.methods()})

    # one PS reporting versions to the master every 2 updates
    ps = ParameterServer(parse_ps_args([
        "--port", "0", "--ps_id", "0", "--num_ps_pods", "1",
        "--opt_type", "sgd", "--opt_args", "learning_rate=0.01",
        "--device", "cpu", "--evaluation_steps", "2",
    ]))
    port = ps.start()
    ps._master_client = MasterClient(f"127.0.0.1:{server.port}", worker_id=-1)
    ps.servicer._master_client = ps._master_client
    try:
        mc = MasterClient(f"127.0.0.1:{server.port}", worker_id=0)
        trainer = ParameterServerTrainer(
            spec, PSClient([f"127.0.0.1:{port}"]), device="cpu"
        )
        worker = Worker(0, mc, trainer, data_reader=reader, spec=spec,
                        minibatch_size=16)
        worker.run()
        assert tm.finished()
        # evaluation ran and produced metrics
        assert "accuracy" in ev.latest_result, ev.latest_result
    finally:
        server.stop(0)
        ps.server.stop(0)
