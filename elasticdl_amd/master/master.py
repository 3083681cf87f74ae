"""Master composition root.

Rebuild of elasticdl/python/master/master.py:32-233 +
elasticdl_job_service.py:57-164: conditionally creates the task manager,
pod manager (local subprocesses or Kubernetes), elastic rendezvous server
(AllReduce), evaluation service and gRPC servicer; wires pod-event
callbacks; builds the worker/PS command lines; polls until all workers
exit or a stop is requested.
"""

import os
import sys
import threading
import time
from typing import List, Optional

from elasticdl_amd.common.args import (
    function_names_from_args,
    parse_envs,
    parse_model_params,
    populated_envs,
)
from elasticdl_amd.common.constants import DistributionStrategy
from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.common.rpc import start_server
from elasticdl_amd.master.evaluation_service import EvaluationService
from elasticdl_amd.master.local_runner import LocalProcessManager
from elasticdl_amd.master.pod_event_callbacks import (
    JobFailureCallback,
    RendezvousServiceRefreshCallback,
    TaskRescheduleCallback,
)
from elasticdl_amd.master.rendezvous import ElasticRendezvousServer
from elasticdl_amd.master.servicer import MasterServicer
from elasticdl_amd.master.task_manager import TaskManager
from elasticdl_amd.utils.model_utils import get_model_spec, get_optimizer_info


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("0.0.0.0", 0))
        return s.getsockname()[1]


class Master:
    def __init__(self, args, k8s_client=None):
        self.args = args
        self._k8s_client = k8s_client  # test injection (fake CoreV1 API)
        self.stopped = threading.Event()
        self.exit_code = 0

        if getattr(args, "log_level", ""):
            import logging

            logging.getLogger("elasticdl_amd").setLevel(
                args.log_level.upper())
        spec = get_model_spec(args.model_def, parse_model_params(args.model_params),
                              model_zoo=getattr(args, "model_zoo", ""),
                              function_names=function_names_from_args(args))
        self.spec = spec

        # ---- data shards
        reader_params = parse_model_params(
            getattr(args, "data_reader_params", ""))

        def _make_reader(origin: str):
            """Same resolution chain for train/validation/prediction data:
            zoo custom_data_reader > generic synthetic:<n> > file factory."""
            if spec.data_reader_fn is not None:
                from elasticdl_amd.data.reader import call_data_reader_fn

                return call_data_reader_fn(
                    spec.data_reader_fn, origin, reader_params)
            from elasticdl_amd.data.reader import (
                create_data_reader,
                synthetic_reader_from_spec,
            )

            return synthetic_reader_from_spec(spec, origin) \
                or create_data_reader(origin, **reader_params)

        reader = None
        training_shards = evaluation_shards = None
        if args.training_data or spec.data_reader_fn is not None:
            reader = _make_reader(args.training_data)
            training_shards = reader.create_shards()
        if args.validation_data:
            evaluation_shards = _make_reader(
                args.validation_data).create_shards()

        prediction_shards = None
        if args.prediction_data:
            prediction_shards = _make_reader(
                args.prediction_data).create_shards()

        # job type derivation (reference: elasticdl_job_service.py:32-54)
        self.job_type = args.job_type or (
            "predict" if prediction_shards and not training_shards
            else "evaluate" if evaluation_shards and not training_shards
            else "train"
        )

        records_per_task = args.minibatch_size * args.num_minibatches_per_task
        self.task_manager = TaskManager(
            training_shards=training_shards if self.job_type == "train" else None,
            evaluation_shards=evaluation_shards,
            prediction_shards=prediction_shards,
            records_per_task=records_per_task,
            num_epochs=args.num_epochs,
            max_step=args.max_step,
            shuffle=args.shuffle,
            shuffle_shards=args.shuffle_shards,
            task_timeout_sec=args.task_timeout_sec,
            task_fault_tolerance=getattr(args, "task_fault_tolerance", True),
            relaunch_timeout_worker=getattr(
                args, "relaunch_timeout_worker", True),
        )
        if self.job_type == "evaluate":
            self.task_manager.create_evaluation_tasks(model_version=0)
        elif self.job_type == "predict":
            self.task_manager.create_prediction_tasks()
        if args.output:
            self.task_manager.enable_train_end_callback()
        if args.checkpoint_dir_for_init:
            from elasticdl_amd.utils.save_utils import latest_valid_version

            v = latest_valid_version(args.checkpoint_dir_for_init)
            if v is not None:
                self.task_manager.set_completed_steps(v)

        # ---- rendezvous (allreduce only)
        self.rendezvous_server: Optional[ElasticRendezvousServer] = None
        if args.distribution_strategy == DistributionStrategy.ALLREDUCE:
            self.rendezvous_server = ElasticRendezvousServer("0.0.0.0")

        # ---- evaluation
        self.evaluation_service = EvaluationService(
            self.task_manager,
            evaluation_steps=args.evaluation_steps,
            metrics_fn=spec.eval_metrics_fn,
        )

        self.port = args.port or _free_port()
        # In k8s mode the address handed to worker/PS pods must be routable
        # from OTHER nodes: the master pod's IP (MY_POD_IP is injected into
        # every elasticdl pod spec, reference elasticdl_job_service.py:65).
        # Local mode keeps loopback.
        if args.pod_manager == "k8s":
            master_ip = os.environ.get("MY_POD_IP", "127.0.0.1")
        else:
            master_ip = "127.0.0.1"
        self.master_addr = f"{master_ip}:{self.port}"

        # ---- pod manager
        self.pod_manager = None
        if args.pod_manager == "local" and args.num_workers > 0:
            self.pod_manager = self._create_local_pod_manager()
        elif args.pod_manager == "k8s":
            self.pod_manager = self._create_k8s_pod_manager()

        self.servicer = MasterServicer(
            self.task_manager,
            rendezvous_server=self.rendezvous_server,
            evaluation_service=self.evaluation_service,
            pod_manager=self.pod_manager,
        )
        self.server = None

    # ------------------------------------------------------------ commands
    def worker_command(self, worker_id: int) -> List[str]:
        a = self.args
        if getattr(a, "job_command", ""):
            # SDK-style job (reference pod_manager.py:327-380): the worker
            # pod runs the user's own command; coordinates come from the
            # EDL_* env the pod managers inject.
            return ["bash", "-c", a.job_command]
        cmd = [
            sys.executable, "-m", "elasticdl_amd.worker.main",
            "--master_addr", self.master_addr,
            "--worker_id", str(worker_id),
            "--model_def", a.model_def,
            "--model_zoo", a.model_zoo,
            "--model_params", a.model_params,
            "--distribution_strategy", a.distribution_strategy,
            "--minibatch_size", str(a.minibatch_size),
            "--get_model_steps", str(a.get_model_steps),
            "--training_data", a.training_data,
            "--validation_data", a.validation_data,
            "--prediction_data", a.prediction_data,
            "--data_reader_params", a.data_reader_params,
            "--device", a.device,
            "--log_loss_steps", str(a.log_loss_steps),
        ]
        if getattr(a, "log_level", ""):
            cmd += ["--log_level", a.log_level]
        for fn_flag in ("loss", "optimizer", "feed", "eval_metrics_fn",
                        "callbacks", "custom_data_reader"):
            v = getattr(a, fn_flag, "")
            if v:
                cmd += [f"--{fn_flag}", v]
        if a.output:
            cmd += ["--output", a.output]
        if self.ps_addrs:
            cmd += ["--ps_addrs", ",".join(self.ps_addrs)]
        return cmd

    def ps_command(self, ps_id: int) -> List[str]:
        a = self.args
        opt_type, opt_args = get_optimizer_info(
            self.spec.optimizer_fn(None)
            if _optimizer_takes_none(self.spec)
            else self.spec.optimizer_fn(self.spec.build_model())
        )
        port = self._ps_ports[ps_id]
        return [
            sys.executable, "-m", "elasticdl_amd.ps.server",
            "--port", str(port),
            "--ps_id", str(ps_id),
            "--num_ps_pods", str(a.num_ps_pods),
            "--num_workers", str(a.num_workers),
            "--opt_type", opt_type,
            "--opt_args", opt_args,
            "--use_async", str(a.use_async),
            "--grads_to_wait", str(a.grads_to_wait),
            "--lr_staleness_modulation", str(a.lr_staleness_modulation),
            "--sync_version_tolerance", str(a.sync_version_tolerance),
            "--evaluation_steps", str(a.evaluation_steps),
            "--checkpoint_dir", a.checkpoint_dir,
            "--checkpoint_steps", str(a.checkpoint_steps),
            "--keep_checkpoint_max", str(a.keep_checkpoint_max),
            "--checkpoint_dir_for_init", a.checkpoint_dir_for_init,
            "--embedding_max_rows", str(a.embedding_max_rows),
            "--device", a.device,
            "--master_addr", self.master_addr,
        ]

    def _create_local_pod_manager(self) -> LocalProcessManager:
        a = self.args
        self._ps_ports = [_free_port() for _ in range(a.num_ps_pods)]
        self.ps_addrs = [f"127.0.0.1:{p}" for p in self._ps_ports]
        mgr = LocalProcessManager(
            master_addr=self.master_addr,
            worker_command=self.worker_command,
            ps_command=self.ps_command,
            num_workers=a.num_workers,
            num_ps=a.num_ps_pods,
            relaunch_on_worker_failure=getattr(
                a, "relaunch_on_worker_failure", 3
            ),
            log_dir=os.path.join(a.checkpoint_dir or "/tmp/edl", "logs"),
            user_envs={
                **populated_envs(getattr(a, "populate_env_names", "")),
                **parse_envs(getattr(a, "envs", "")),
            },
        )
        mgr.ps_addrs = self.ps_addrs
        mgr.add_pod_event_callback(TaskRescheduleCallback(self.task_manager))
        if self.rendezvous_server is not None:
            mgr.add_pod_event_callback(
                RendezvousServiceRefreshCallback(self.rendezvous_server)
            )
        if a.num_ps_pods > 0:
            mgr.add_pod_event_callback(JobFailureCallback(self))
        return mgr

    def _create_k8s_pod_manager(self):
        from elasticdl_amd.master.k8s_client import PS_SERVICE_PORT
        from elasticdl_amd.master.pod_manager import PodManager

        mgr = PodManager(self.args, self, k8s_client=self._k8s_client)
        # Every PS pod serves on the same fixed port behind its own
        # per-pod Service; workers address PS shards by service DNS
        # (reference pod_manager.py:269 + k8s_client.py:126-128). This is
        # what makes cross-node PS discovery work — pod IPs are neither
        # known at command-build time nor stable across relaunches.
        self._ps_ports = [PS_SERVICE_PORT] * self.args.num_ps_pods
        self.ps_addrs = [
            mgr.k8s.get_ps_service_address(i)
            for i in range(self.args.num_ps_pods)
        ]
        mgr.ps_addrs = self.ps_addrs
        mgr.add_pod_event_callback(TaskRescheduleCallback(self.task_manager))
        if self.rendezvous_server is not None:
            mgr.add_pod_event_callback(
                RendezvousServiceRefreshCallback(self.rendezvous_server)
            )
        if self.args.num_ps_pods > 0:
            mgr.add_pod_event_callback(JobFailureCallback(self))
        return mgr

    # ------------------------------------------------------------ lifecycle
    def prepare(self) -> None:
        if self.rendezvous_server is not None:
            self.rendezvous_server.start()
        self.server = start_server(
            f"0.0.0.0:{self.port}", {"Master": self.servicer.methods()}
        )
        self.task_manager.start()
        if self.pod_manager is not None:
            self.task_manager.register_task_timeout_callback(
                getattr(self.pod_manager, "kill_worker", lambda wid: None)
            )
            self.pod_manager.start()
            if self.args.num_ps_pods > 0:
                self.pod_manager.start_parameter_servers()
            self.pod_manager.start_workers()
        logger.info("Master serving on %s", self.master_addr)

    def run(self, poll_interval: float = 0.5) -> int:
        try:
            while not self.stopped.wait(poll_interval):
                if self.pod_manager is not None:
                    if self.pod_manager.all_workers_exited():
                        if self.pod_manager.all_workers_failed():
                            logger.error("All workers failed")
                            self.exit_code = 1
                        break
                elif self.task_manager.finished():
                    break
        finally:
            self.cleanup()
        # a job whose data was entirely dropped (every task exhausted its
        # retries) must not report success — reference counts failed
        # records (task_manager.py:71-93); we additionally fail the job
        # when NOTHING succeeded
        tm = self.task_manager
        if tm.failed_records:
            logger.warning(
                "Job dropped %d records after task retries (completed %d)",
                tm.failed_records, tm.counts()["completed_records"],
            )
            if tm.counts()["completed_records"] == 0:
                logger.error("All task shards failed; marking job failed")
                self.exit_code = 1
        return self.exit_code

    def request_stop(self, success: bool = True) -> None:
        self.exit_code = 0 if success else 1
        self.stopped.set()

    def cleanup(self) -> None:
        self.task_manager.stop()
        if self.pod_manager is not None:
            self.pod_manager.stop()
            # advertise final job status on the master pod for external
            # observers (reference: pod_manager.py:444-448)
            k8s = getattr(self.pod_manager, "k8s", None)
            if k8s is not None:
                k8s.patch_labels_to_pod(
                    k8s.get_master_pod_name(),
                    {"status": "Finished" if self.exit_code == 0 else "Failed"},
                )
        if self.server is not None:
            self.server.stop(1)

    ps_addrs: List[str] = []
    _ps_ports: List[int] = []


def _optimizer_takes_none(spec) -> bool:
    import inspect

    try:
        sig = inspect.signature(spec.optimizer_fn)
        params = list(sig.parameters.values())
        return all(
            p.default is not inspect.Parameter.empty for p in params
        )
    except (TypeError, ValueError):
        return False
