"""Minimal stand-ins for the ``kubernetes.client`` V1* spec types.

The master builds pod/service specs through these constructors; when the
real ``kubernetes`` package is installed it is used directly (see
``k8s_client.k8s_types()``), and in offline environments (CI, unit tests
with a fake CoreV1Api) these attribute-bag equivalents keep the full
spec-building + event-watch code path testable without a cluster SDK.
Only the constructor keywords the framework uses are modeled.
"""


class _Spec:
    _fields: tuple = ()

    def __init__(self, **kw):
        for f in self._fields:
            setattr(self, f, kw.pop(f, None))
        if kw:
            raise TypeError(
                f"{type(self).__name__} got unexpected fields {sorted(kw)}"
            )

    def __repr__(self):
        pairs = ", ".join(
            f"{f}={getattr(self, f)!r}"
            for f in self._fields
            if getattr(self, f) is not None
        )
        return f"{type(self).__name__}({pairs})"


class V1ObjectMeta(_Spec):
    _fields = ("name", "labels", "annotations", "owner_references",
               "namespace", "uid")


class V1EnvVar(_Spec):
    _fields = ("name", "value", "value_from")


class V1EnvVarSource(_Spec):
    _fields = ("field_ref",)


class V1ObjectFieldSelector(_Spec):
    _fields = ("field_path",)


class V1ResourceRequirements(_Spec):
    _fields = ("requests", "limits")


class V1VolumeMount(_Spec):
    _fields = ("name", "mount_path")


class V1Volume(_Spec):
    _fields = ("name", "persistent_volume_claim", "host_path")


class V1PersistentVolumeClaimVolumeSource(_Spec):
    _fields = ("claim_name",)


class V1HostPathVolumeSource(_Spec):
    _fields = ("path",)


class V1Container(_Spec):
    _fields = ("name", "image", "command", "args", "resources", "env",
               "volume_mounts", "image_pull_policy")


class V1PodSpec(_Spec):
    _fields = ("containers", "restart_policy", "priority_class_name",
               "volumes", "termination_grace_period_seconds",
               "tolerations", "affinity", "node_selector")


class V1OwnerReference(_Spec):
    _fields = ("api_version", "kind", "name", "uid",
               "block_owner_deletion", "controller")


class V1Pod(_Spec):
    _fields = ("api_version", "kind", "metadata", "spec", "status")


class V1ServicePort(_Spec):
    _fields = ("port", "target_port")


class V1ServiceSpec(_Spec):
    _fields = ("ports", "selector", "type")


class V1Service(_Spec):
    _fields = ("api_version", "kind", "metadata", "spec")


class V1DeleteOptions(_Spec):
    _fields = ("grace_period_seconds",)
