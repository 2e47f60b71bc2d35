"""Kubernetes scheduler via Volcano gang scheduling (parity:
torchx/schedulers/kubernetes_scheduler.py).

Generates a ``batch.volcano.sh/v1alpha1`` Job with one task per replica
(macros vary per replica), gang-scheduled with ``minAvailable`` (elastic
via role.min_replicas).  MI355X-native resource wiring: GPUs are requested
as ``amd.com/gpu`` (the AMD k8s device plugin's resource name) and pods get
a memory-backed /dev/shm emptyDir sized for RCCL transports.  Rank-0
discovery uses Volcano's svc plugin env ``VC_{role}_0_HOSTS``; retry
policies map to Volcano lifecycle events.  The request is built as plain
dicts so it is fully testable without a cluster; the kubernetes SDK is
imported only at schedule/describe time."""

from __future__ import annotations

import json
import logging
import re
from dataclasses import dataclass
from datetime import datetime
from typing import Any, Dict, Iterable, List, Mapping, Optional

import yaml

from torchx_amd.specs import (
    AppDef,
    AppDryRunInfo,
    AppState,
    BindMount,
    DeviceMount,
    ReplicaStatus,
    RetryPolicy,
    Role,
    RoleStatus,
    VolumeMount,
    macros,
    runopts,
)
from torchx_amd.specs.overlays import OverlaySpec, apply_overlay

from torchx_amd.workspace.docker_workspace import DockerWorkspaceMixin

from .api import DescribeAppResponse, ListAppResponse, Scheduler, Stream
from .ids import make_unique

log = logging.getLogger(__name__)

# overlay surface: users patch the generated pod via
# set_overlay(role, "kubernetes", "V1Pod", {...}); fields owned by the Role
# are rejected at write time
V1POD_OVERLAY = OverlaySpec(
    "kubernetes", "V1Pod", blocklist=("command", "env", "image")
)

# reserve headroom for k8s daemons (parity: RESERVED_MILLICPU/MEMMB)
RESERVED_MILLICPU = 100
RESERVED_MEMMB = 1024

RETRY_POLICIES: Dict[str, List[Dict[str, str]]] = {
    str(RetryPolicy.REPLICA): [],
    str(RetryPolicy.APPLICATION): [
        {"event": "PodEvicted", "action": "RestartJob"},
        {"event": "PodFailed", "action": "RestartJob"},
    ],
    str(RetryPolicy.ROLE): [
        {"event": "PodEvicted", "action": "RestartTask"},
        {"event": "PodFailed", "action": "RestartTask"},
    ],
}

JOB_STATE: Dict[str, AppState] = {
    "Pending": AppState.PENDING,
    "Aborting": AppState.RUNNING,
    "Aborted": AppState.CANCELLED,
    "Running": AppState.RUNNING,
    "Restarting": AppState.RUNNING,
    "Completing": AppState.RUNNING,
    "Completed": AppState.SUCCEEDED,
    "Terminating": AppState.RUNNING,
    "Terminated": AppState.FAILED,
    "Failed": AppState.FAILED,
    "Inqueue": AppState.PENDING,
}

LABEL_APP_NAME = "torchx.ai/app-name"
LABEL_ROLE_NAME = "torchx.ai/role-name"
LABEL_REPLICA_ID = "torchx.ai/replica-id"
LABEL_VERSION = "torchx.ai/version"

AMD_GPU_RESOURCE = "amd.com/gpu"


def sanitize_for_k8s(name: str) -> str:
    s = re.sub(r"[^a-z0-9\-]", "-", name.lower()).strip("-")
    return s[:63]


def role_to_pod(name: str, role: Role, service_account: Optional[str],
                image_pull_policy: str = "IfNotPresent") -> Dict[str, Any]:
    """Build the pod template dict for one replica."""
    res = role.resource
    limits: Dict[str, Any] = {}
    requests: Dict[str, Any] = {}
    if res.cpu > 0:
        mcpu = int(res.cpu * 1000)
        limits["cpu"] = f"{mcpu}m"
        requests["cpu"] = f"{max(0, mcpu - RESERVED_MILLICPU)}m"
    if res.memMB > 0:
        limits["memory"] = f"{res.memMB}M"
        requests["memory"] = f"{max(0, res.memMB - RESERVED_MEMMB)}M"
    if res.gpu > 0:
        limits[AMD_GPU_RESOURCE] = res.gpu
        requests[AMD_GPU_RESOURCE] = res.gpu

    volumes: List[Dict[str, Any]] = [
        # RCCL needs generous /dev/shm for its shared-memory transport
        {"name": "dshm", "emptyDir": {"medium": "Memory"}},
    ]
    volume_mounts: List[Dict[str, Any]] = [
        {"name": "dshm", "mountPath": "/dev/shm"},
    ]
    security_context: Dict[str, Any] = {}
    for i, m in enumerate(role.mounts):
        mname = f"mount-{i}"
        if isinstance(m, BindMount):
            volumes.append(
                {"name": mname, "hostPath": {"path": m.src_path}}
            )
            volume_mounts.append(
                {"name": mname, "mountPath": m.dst_path,
                 "readOnly": m.read_only}
            )
        elif isinstance(m, VolumeMount):
            volumes.append(
                {"name": mname,
                 "persistentVolumeClaim": {"claimName": m.src}}
            )
            volume_mounts.append(
                {"name": mname, "mountPath": m.dst_path,
                 "readOnly": m.read_only}
            )
        elif isinstance(m, DeviceMount):
            volumes.append(
                {"name": mname, "hostPath": {"path": m.src_path}}
            )
            volume_mounts.append(
                {"name": mname, "mountPath": m.dst_path,
                 "readOnly": "w" not in m.permissions}
            )
            security_context["privileged"] = True

    container: Dict[str, Any] = {
        "name": name,
        "image": role.image,
        "command": [role.entrypoint, *role.args],
        "env": [{"name": k, "value": v} for k, v in role.env.items()],
        "resources": {"limits": limits, "requests": requests},
        "volumeMounts": volume_mounts,
    }
    if security_context:
        container["securityContext"] = security_context

    spec: Dict[str, Any] = {
        "containers": [container],
        "restartPolicy": "Never",
        "volumes": volumes,
    }
    if service_account:
        spec["serviceAccountName"] = service_account
    node_selector = {
        k: str(v)
        for k, v in res.capabilities.items()
        if k.startswith("node.kubernetes.io/") or k.startswith("kubernetes.io/")
    }
    if node_selector:
        spec["nodeSelector"] = node_selector
    return {"metadata": {}, "spec": spec}


def app_to_resource(app: AppDef, queue: str,
                    service_account: Optional[str],
                    priority_class: Optional[str] = None) -> Dict[str, Any]:
    """AppDef -> Volcano Job custom resource dict."""
    app_id = make_unique(sanitize_for_k8s(app.name))
    tasks: List[Dict[str, Any]] = []
    total_replicas = 0
    min_available = 0
    for role_idx, role in enumerate(app.roles):
        for replica_id in range(role.num_replicas):
            values = macros.Values(
                img_root="",
                app_id=app_id,
                replica_id=str(replica_id),
                rank0_env=f"VC_{sanitize_for_k8s(app.roles[0].name).upper().replace('-', '')}_0_HOSTS",
            )
            replica = values.apply(role)
            if role_idx == 0 and replica_id == 0:
                replica.env["TORCHX_RANK0_HOST"] = "localhost"
            name = sanitize_for_k8s(f"{role.name}-{replica_id}")
            pod = role_to_pod(name, replica, service_account)
            # user pod overlay (reference parity: kubernetes_scheduler.py:164
            # _apply_pod_overlay); command/env/resources come from the Role
            pod_overlay = V1POD_OVERLAY.get(role)
            if pod_overlay:
                apply_overlay(pod, pod_overlay)
            pod["metadata"].setdefault("labels", {}).update(
                {
                    LABEL_APP_NAME: sanitize_for_k8s(app.name),
                    LABEL_ROLE_NAME: sanitize_for_k8s(role.name),
                    LABEL_REPLICA_ID: str(replica_id),
                    LABEL_VERSION: "0.1.0",
                }
            )
            task = {
                "replicas": 1,
                "name": name,
                "template": pod,
                "maxRetry": role.max_retries,
                "policies": RETRY_POLICIES.get(str(role.retry_policy), []),
            }
            min_replicas = role.min_replicas
            if min_replicas is not None:
                # elastic: only the first min_replicas tasks gate the gang
                task["minAvailable"] = 1 if replica_id < min_replicas else 0
            tasks.append(task)
            total_replicas += 1
            if min_replicas is None or replica_id < min_replicas:
                min_available += 1

    job_spec: Dict[str, Any] = {
        "schedulerName": "volcano",
        "queue": queue,
        "tasks": tasks,
        "minAvailable": min_available,
        "maxRetry": min((r.max_retries for r in app.roles), default=0),
        "plugins": {"svc": [], "env": []},
    }
    if priority_class:
        job_spec["priorityClassName"] = priority_class
    return {
        "apiVersion": "batch.volcano.sh/v1alpha1",
        "kind": "Job",
        "metadata": {"name": app_id},
        "spec": job_spec,
    }


@dataclass
class KubernetesJob:
    resource: Dict[str, Any]
    images_to_push: Optional[Any] = None
    namespace: str = "default"

    def __str__(self) -> str:
        return yaml.dump(self.resource, sort_keys=False)


class KubernetesScheduler(DockerWorkspaceMixin, Scheduler[KubernetesJob]):
    def __init__(self, session_name: str, client: Optional[Any] = None) -> None:
        super().__init__("kubernetes", session_name)
        self.__client = client

    def _api(self):
        if self.__client is None:
            from kubernetes import client, config

            try:
                config.load_incluster_config()
            except Exception:  # noqa: BLE001
                config.load_kube_config()
            self.__client = client.ApiClient()
        return self.__client

    def _custom_api(self):
        from kubernetes import client

        return client.CustomObjectsApi(self._api())

    def _core_api(self):
        from kubernetes import client

        return client.CoreV1Api(self._api())

    def run_opts(self) -> runopts:
        opts = runopts()
        opts.add("namespace", type_=str, default="default",
                 help="kubernetes namespace")
        opts.add("queue", type_=str, required=True, help="volcano queue")
        opts.add("service_account", type_=str, default=None,
                 help="pod service account")
        opts.add("priority_class", type_=str, default=None,
                 help="pod priority class")
        opts.add("validate_spec", type_=bool, default=False,
                 help="server-side dry-run validation before submit")
        return opts

    def _submit_dryrun(
        self, app: AppDef, cfg: Mapping[str, Any]
    ) -> AppDryRunInfo[KubernetesJob]:
        # locally-built sha256 images must be pushed to image_repo before
        # the cluster can pull them (reference parity: k8s_scheduler.py:824)
        images_to_push = self.dryrun_push_images(app, dict(cfg))
        resource = app_to_resource(
            app,
            queue=str(cfg.get("queue")),
            service_account=cfg.get("service_account"),
            priority_class=cfg.get("priority_class"),
        )
        job = KubernetesJob(
            resource=resource, images_to_push=images_to_push,
            namespace=str(cfg.get("namespace", "default")),
        )
        return AppDryRunInfo(request=job, fmt=str)

    def schedule(self, dryrun_info: AppDryRunInfo[KubernetesJob]) -> str:
        req = dryrun_info.request
        self.push_images(req.images_to_push)
        api = self._custom_api()
        resp = api.create_namespaced_custom_object(
            group="batch.volcano.sh",
            version="v1alpha1",
            namespace=req.namespace,
            plural="jobs",
            body=req.resource,
        )
        return f"{req.namespace}:{resp['metadata']['name']}"

    def _split(self, app_id: str):
        ns, _, name = app_id.partition(":")
        return ns, name

    def describe(self, app_id: str) -> Optional[DescribeAppResponse]:
        ns, name = self._split(app_id)
        api = self._custom_api()
        try:
            job = api.get_namespaced_custom_object(
                group="batch.volcano.sh", version="v1alpha1", namespace=ns,
                plural="jobs", name=name,
            )
        except Exception as e:  # noqa: BLE001 — duck-typed ApiException so
            # the module works without the kubernetes SDK installed
            # (reference parity: KubernetesSchedulerNoImportTest)
            if getattr(e, "status", None) == 404:
                return None
            raise
        status = job.get("status", {})
        state = JOB_STATE.get(status.get("state", {}).get("phase", ""),
                              AppState.UNKNOWN)
        roles: Dict[str, RoleStatus] = {}
        task_counts = status.get("taskStatusCount", {})
        for task_name, counts in task_counts.items():
            role, _, ridx = task_name.rpartition("-")
            rs = roles.setdefault(role, RoleStatus(role=role))
            phase = next(iter(counts.get("phase", {"Unknown": 1})))
            pod_state = {
                "Pending": AppState.PENDING,
                "Running": AppState.RUNNING,
                "Succeeded": AppState.SUCCEEDED,
                "Failed": AppState.FAILED,
            }.get(phase, AppState.UNKNOWN)
            rs.replicas.append(
                ReplicaStatus(
                    id=int(ridx) if ridx.isdigit() else 0,
                    state=pod_state, role=role,
                )
            )
        return DescribeAppResponse(
            app_id=app_id, state=state, roles_statuses=list(roles.values())
        )

    def list(self) -> List[ListAppResponse]:
        api = self._custom_api()
        jobs = api.list_namespaced_custom_object(
            group="batch.volcano.sh", version="v1alpha1",
            namespace="default", plural="jobs",
        )
        out = []
        for job in jobs.get("items", []):
            name = job["metadata"]["name"]
            phase = job.get("status", {}).get("state", {}).get("phase", "")
            out.append(
                ListAppResponse(
                    app_id=f"default:{name}",
                    state=JOB_STATE.get(phase, AppState.UNKNOWN),
                )
            )
        return out

    def _cancel_existing(self, app_id: str) -> None:
        ns, name = self._split(app_id)
        api = self._custom_api()
        # abort (preserves the job spec for inspection), parity :900
        api.patch_namespaced_custom_object(
            group="batch.volcano.sh", version="v1alpha1", namespace=ns,
            plural="jobs", name=name,
            body={"spec": {"suspend": True}},
        )
        api.delete_namespaced_custom_object(
            group="batch.volcano.sh", version="v1alpha1", namespace=ns,
            plural="jobs", name=name,
        )

    def log_iter(
        self,
        app_id: str,
        role_name: str,
        k: int = 0,
        regex: Optional[str] = None,
        since: Optional[datetime] = None,
        until: Optional[datetime] = None,
        should_tail: bool = False,
        streams: Optional[Stream] = None,
    ) -> Iterable[str]:
        ns, name = self._split(app_id)
        pod_name = f"{name}-{sanitize_for_k8s(role_name)}-{k}-0"
        core = self._core_api()
        if should_tail:
            from kubernetes import watch

            w = watch.Watch()
            lines = w.stream(
                core.read_namespaced_pod_log, name=pod_name, namespace=ns
            )
        else:
            text = core.read_namespaced_pod_log(name=pod_name, namespace=ns)
            lines = text.splitlines()
        if regex:
            from .api import filter_regex

            lines = filter_regex(regex, lines)
        return lines


def create_scheduler(session_name: str, **kwargs: Any) -> KubernetesScheduler:
    return KubernetesScheduler(session_name=session_name,
                               client=kwargs.get("client"))
