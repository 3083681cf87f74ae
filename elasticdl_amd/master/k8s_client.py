"""Kubernetes client wrapper: pod CRUD + label patch + event watch.

Rebuild of the reference's two k8s clients
(elasticdl_client/common/k8s_client.py:220-410 — pod/service CRUD,
labels, owner references; elasticdl/python/common/k8s_client.py:41-111 —
the master-side event-watch thread). The ``kubernetes`` package import is
deferred so the rest of the framework works without it; tests inject a
fake core API.

Labels (reference: k8s_client.py:29-33):
    elasticdl-job-name, elasticdl-replica-type, elasticdl-replica-index
"""

import threading
import traceback
from typing import Callable, Dict, List, Optional

from elasticdl_amd.common.log_utils import default_logger as logger

ELASTICDL_JOB_KEY = "elasticdl-job-name"
ELASTICDL_REPLICA_TYPE_KEY = "elasticdl-replica-type"
ELASTICDL_REPLICA_INDEX_KEY = "elasticdl-replica-index"


def k8s_types():
    """The ``kubernetes.client`` module when installed, else the offline
    attribute-bag equivalents (k8s_types.py) so spec building stays
    testable without the cluster SDK."""
    try:
        from kubernetes import client as k8s

        return k8s
    except ImportError:
        from elasticdl_amd.master import k8s_types as k8s

        return k8s

# fixed in-pod service ports (reference: common/k8s_client.py:29-30) —
# every PS pod serves on the same port behind its own per-pod Service,
# so cross-pod addresses are stable DNS names independent of pod IPs
PS_SERVICE_PORT = 2222
WORKER_SERVICE_PORT = 3333


def parse_resource(spec: str) -> Dict[str, str]:
    """'cpu=4,memory=8192Mi,amd.com/gpu=1' -> k8s resource dict
    (reference: elasticdl_client/common/k8s_resource.py)."""
    out: Dict[str, str] = {}
    for kv in (spec or "").split(","):
        kv = kv.strip()
        if not kv:
            continue
        k, _, v = kv.partition("=")
        k = k.strip()
        if k == "gpu":
            k = "amd.com/gpu"
        out[k] = v.strip()
    return out


def parse_volume(spec: str) -> List[Dict[str, str]]:
    """'claim_name=x,mount_path=/data;...' -> volume dicts
    (reference: k8s_volume.py)."""
    out = []
    for part in (spec or "").split(";"):
        part = part.strip()
        if not part:
            continue
        vol = {}
        for kv in part.split(","):
            k, _, v = kv.partition("=")
            vol[k.strip()] = v.strip()
        out.append(vol)
    return out


class ClusterSpec:
    """Bespoke-cluster pod/service mutation hook (reference:
    elasticdl_client/common/k8s_client.py:106-219). Two forms, composable:

    - ``spec``: a python module/file exposing ``patch_pod(pod, pod_type)``
      (and optionally ``patch_service(service)``);
    - ``spec_json``: a JSON dict with declarative additions —
      ``pod_spec`` (all pods), ``master_spec``/``ps_spec``/``worker_spec``
      (per type), ``service_spec``; each may set labels, annotations,
      env (name/value pairs), tolerations, affinity, node_selector.
    """

    def __init__(self, spec: str = "", spec_json: str = ""):
        self._mod = None
        self._json = None
        if spec:
            from elasticdl_amd.utils.model_utils import load_module

            self._mod = load_module(spec)
        if spec_json:
            import json

            self._json = json.loads(spec_json)

    @staticmethod
    def _apply_pod_spec(pod, d: dict):
        meta = pod.metadata
        if "labels" in d:
            meta.labels = {**(meta.labels or {}), **d["labels"]}
        if "annotations" in d:
            meta.annotations = {
                **(getattr(meta, "annotations", None) or {}),
                **d["annotations"],
            }
        k8s = k8s_types()
        for e in d.get("env", []):
            for c in pod.spec.containers:
                c.env = list(c.env or [])
                c.env.append(k8s.V1EnvVar(name=e["name"], value=e["value"]))
        if "tolerations" in d:
            pod.spec.tolerations = d["tolerations"]
        if "affinity" in d:
            pod.spec.affinity = d["affinity"]
        if "node_selector" in d:
            pod.spec.node_selector = d["node_selector"]
        return pod

    def patch_pod(self, pod, pod_type: str):
        if self._mod is not None and hasattr(self._mod, "patch_pod"):
            pod = self._mod.patch_pod(pod, pod_type) or pod
        if self._json is not None:
            for key in ("pod_spec", f"{pod_type}_spec"):
                if key in self._json:
                    pod = self._apply_pod_spec(pod, self._json[key])
        return pod

    def patch_service(self, service):
        if self._mod is not None and hasattr(self._mod, "patch_service"):
            service = self._mod.patch_service(service) or service
        if self._json is not None and "service_spec" in self._json:
            d = self._json["service_spec"]
            meta = service.metadata
            if "labels" in d:
                meta.labels = {**(getattr(meta, "labels", None) or {}),
                               **d["labels"]}
            if "annotations" in d:
                meta.annotations = {
                    **(getattr(meta, "annotations", None) or {}),
                    **d["annotations"],
                }
        return service


class Client:
    def __init__(
        self,
        namespace: str,
        job_name: str,
        image_name: str = "",
        core_api=None,
        force_use_kube_config_file: bool = False,
    ):
        self.namespace = namespace
        self.job_name = job_name
        self.image_name = image_name
        if core_api is not None:
            self.client = core_api
        else:
            self.client = self._create_core_api(force_use_kube_config_file)

    @staticmethod
    def _create_core_api(use_kube_config: bool):
        from kubernetes import client, config

        try:
            if use_kube_config:
                config.load_kube_config()
            else:
                config.load_incluster_config()
        except Exception:  # noqa: BLE001 - fall back to kubeconfig
            config.load_kube_config()
        return client.CoreV1Api()

    # ------------------------------------------------------------- naming
    def get_master_pod_name(self) -> str:
        return f"elasticdl-{self.job_name}-master"

    def get_pod_name(self, pod_type: str, index: int) -> str:
        return f"elasticdl-{self.job_name}-{pod_type}-{index}"

    def get_service_name(self, pod_type: str, index: int) -> str:
        """Service name == pod name (reference: get_ps_service_name)."""
        return self.get_pod_name(pod_type, index)

    def get_service_address(self, pod_type: str, index: int,
                            port: int) -> str:
        """Cluster-DNS address of a replica's Service
        (reference: common/k8s_client.py:113-114)."""
        return f"{self.get_service_name(pod_type, index)}." \
               f"{self.namespace}.svc:{port}"

    def get_ps_service_address(self, ps_id: int) -> str:
        return self.get_service_address("ps", ps_id, PS_SERVICE_PORT)

    # --------------------------------------------------------------- CRUD
    def create_pod(self, pod) -> bool:
        try:
            self.client.create_namespaced_pod(self.namespace, pod)
            return True
        except Exception:  # noqa: BLE001
            logger.warning("create_pod failed:\n%s", traceback.format_exc())
            return False

    def delete_pod(self, pod_name: str) -> bool:
        try:
            self.client.delete_namespaced_pod(
                pod_name,
                self.namespace,
                body=k8s_types().V1DeleteOptions(grace_period_seconds=0),
            )
            return True
        except Exception:  # noqa: BLE001
            logger.warning("delete_pod(%s) failed", pod_name)
            return False

    def get_pod(self, pod_name: str):
        try:
            return self.client.read_namespaced_pod(pod_name, self.namespace)
        except Exception:  # noqa: BLE001
            return None

    def create_service(self, service) -> bool:
        try:
            self.client.create_namespaced_service(self.namespace, service)
            return True
        except Exception:  # noqa: BLE001
            logger.warning("create_service failed:\n%s",
                           traceback.format_exc())
            return False

    def patch_service(self, service_name: str, service) -> bool:
        try:
            self.client.patch_namespaced_service(
                service_name, self.namespace, service
            )
            return True
        except Exception:  # noqa: BLE001
            logger.warning("patch_service(%s) failed", service_name)
            return False

    def get_service(self, service_name: str):
        try:
            return self.client.read_namespaced_service(
                service_name, self.namespace
            )
        except Exception:  # noqa: BLE001
            return None

    def build_service_spec(
        self,
        pod_type: str,
        index: int,
        port: int,
        target_port: int = 0,
        owner_pod=None,
        selector_index: Optional[int] = None,
    ):
        """V1Service selecting one replica by (job, type, index) labels
        (reference: common/k8s_client.py:275-311). ``selector_index`` lets
        a relaunched worker be patched behind the original service name."""
        k8s = k8s_types()

        labels = {
            "app": "elasticdl",
            ELASTICDL_JOB_KEY: self.job_name,
            ELASTICDL_REPLICA_TYPE_KEY: pod_type,
            ELASTICDL_REPLICA_INDEX_KEY: str(index),
        }
        selector = dict(labels)
        if selector_index is not None:
            selector[ELASTICDL_REPLICA_INDEX_KEY] = str(selector_index)
        owner_refs = None
        if owner_pod is not None:
            owner_refs = [
                k8s.V1OwnerReference(
                    api_version="v1",
                    kind="Pod",
                    name=owner_pod.metadata.name,
                    uid=owner_pod.metadata.uid,
                    block_owner_deletion=True,
                    controller=True,
                )
            ]
        return k8s.V1Service(
            api_version="v1",
            kind="Service",
            metadata=k8s.V1ObjectMeta(
                name=self.get_service_name(pod_type, index),
                labels=labels,
                # at least one annotation so cluster-spec hooks can extend
                annotations=dict(labels),
                owner_references=owner_refs,
                namespace=self.namespace,
            ),
            spec=k8s.V1ServiceSpec(
                ports=[
                    k8s.V1ServicePort(
                        port=port, target_port=target_port or port
                    )
                ],
                selector=selector,
            ),
        )

    def patch_labels_to_pod(self, pod_name: str, labels: Dict[str, str]):
        try:
            return self.client.patch_namespaced_pod(
                pod_name, self.namespace, {"metadata": {"labels": labels}}
            )
        except Exception:  # noqa: BLE001
            logger.warning("patch labels on %s failed", pod_name)
            return None

    # -------------------------------------------------------------- specs
    def build_pod_spec(
        self,
        pod_name: str,
        pod_type: str,
        index: int,
        command: List[str],
        resource_requests: str,
        resource_limits: str = "",
        priority_class: str = "",
        envs: Optional[Dict[str, str]] = None,
        volumes: str = "",
        image_pull_policy: str = "IfNotPresent",
        restart_policy: str = "Never",
        owner_pod=None,
    ):
        """V1Pod with the elasticdl labels + owner reference to the master
        pod (reference: k8s_client.py:283-298)."""
        k8s = k8s_types()

        env = [k8s.V1EnvVar(name=k, value=str(v)) for k, v in (envs or {}).items()]
        env.append(
            k8s.V1EnvVar(
                name="MY_POD_IP",
                value_from=k8s.V1EnvVarSource(
                    field_ref=k8s.V1ObjectFieldSelector(field_path="status.podIP")
                ),
            )
        )
        requests = parse_resource(resource_requests)
        limits = parse_resource(resource_limits) or requests
        volume_mounts = []
        pod_volumes = []
        for i, vol in enumerate(parse_volume(volumes)):
            name = vol.get("name", f"edl-volume-{i}")
            if "claim_name" in vol:
                pod_volumes.append(
                    k8s.V1Volume(
                        name=name,
                        persistent_volume_claim=(
                            k8s.V1PersistentVolumeClaimVolumeSource(
                                claim_name=vol["claim_name"]
                            )
                        ),
                    )
                )
            elif "host_path" in vol:
                pod_volumes.append(
                    k8s.V1Volume(
                        name=name,
                        host_path=k8s.V1HostPathVolumeSource(
                            path=vol["host_path"]
                        ),
                    )
                )
            volume_mounts.append(
                k8s.V1VolumeMount(name=name, mount_path=vol["mount_path"])
            )
        container = k8s.V1Container(
            name="main",
            image=self.image_name,
            command=command,
            resources=k8s.V1ResourceRequirements(
                requests=requests, limits=limits
            ),
            env=env,
            volume_mounts=volume_mounts or None,
            image_pull_policy=image_pull_policy,
        )
        spec = k8s.V1PodSpec(
            containers=[container],
            restart_policy=restart_policy,
            priority_class_name=priority_class or None,
            volumes=pod_volumes or None,
        )
        owner_refs = None
        if owner_pod is not None:
            owner_refs = [
                k8s.V1OwnerReference(
                    api_version="v1",
                    kind="Pod",
                    name=owner_pod.metadata.name,
                    uid=owner_pod.metadata.uid,
                    block_owner_deletion=True,
                    controller=True,
                )
            ]
        return k8s.V1Pod(
            api_version="v1",
            kind="Pod",
            metadata=k8s.V1ObjectMeta(
                name=pod_name,
                labels={
                    "app": "elasticdl",
                    ELASTICDL_JOB_KEY: self.job_name,
                    ELASTICDL_REPLICA_TYPE_KEY: pod_type,
                    ELASTICDL_REPLICA_INDEX_KEY: str(index),
                },
                owner_references=owner_refs,
            ),
            spec=spec,
        )

    # -------------------------------------------------------------- watch
    def start_watch(self, event_callback: Callable, periodic_callback:
                    Optional[Callable] = None, interval: float = 15.0):
        """Label-selected pod event watch in a daemon thread
        (reference: common/k8s_client.py:92-106) plus an optional periodic
        callback (retry pod creation etc.)."""

        selector = f"{ELASTICDL_JOB_KEY}={self.job_name}"

        def watch_loop():
            while True:
                try:
                    if hasattr(self.client, "stream_pod_events"):
                        # fake/offline API: it owns the event stream
                        # (returning ends the watch — no reconnect loop)
                        for event in self.client.stream_pod_events(
                            self.namespace, label_selector=selector
                        ):
                            event_callback(event)
                        return
                    from kubernetes import watch

                    stream = watch.Watch().stream(
                        self.client.list_namespaced_pod,
                        self.namespace,
                        label_selector=selector,
                    )
                    for event in stream:
                        event_callback(event)
                except Exception:  # noqa: BLE001 - watch reconnects
                    logger.warning("k8s watch reconnecting:\n%s",
                                   traceback.format_exc())
                import time

                time.sleep(2)

        t = threading.Thread(target=watch_loop, name="k8s-watch", daemon=True)
        t.start()
        if periodic_callback is not None:
            def periodic_loop():
                import time

                while True:
                    time.sleep(interval)
                    try:
                        periodic_callback()
                    except Exception:  # noqa: BLE001
                        logger.warning("periodic callback failed")

            threading.Thread(
                target=periodic_loop, name="k8s-periodic", daemon=True
            ).start()
        return t
