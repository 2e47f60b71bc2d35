"""Docker scheduler (``local_docker``) — one container per replica on a
shared bridge network (parity: torchx/schedulers/docker_scheduler.py).

MI355X-native GPU wiring: ROCm containers get the KFD compute node and DRI
render nodes (``--device /dev/kfd --device /dev/dri``) plus
``HIP_VISIBLE_DEVICES`` pinning — not CUDA DeviceRequests.  Rank-0
discovery: ``TORCHX_RANK0_HOST`` is the rank-0 container's name, resolvable
over the bridge network's DNS."""

from __future__ import annotations

import json
import logging
import os
from dataclasses import dataclass, field
from datetime import datetime
from typing import Any, Dict, Iterable, List, Mapping, Optional

from torchx_amd.specs import (
    AppDef,
    AppDryRunInfo,
    AppState,
    BindMount,
    DeviceMount,
    ReplicaStatus,
    RoleStatus,
    VolumeMount,
    is_terminal,
    macros,
    runopts,
)

from torchx_amd.workspace.docker_workspace import DockerWorkspaceMixin

from .api import (
    DescribeAppResponse, ListAppResponse, Scheduler, Stream, StructuredOpts,
)
from .devices import get_device_mounts
from .ids import make_unique

log = logging.getLogger(__name__)

NETWORK_NAME = "torchx"
LABEL_APP_ID = "torchx.ai/app-id"
LABEL_ROLE = "torchx.ai/role"
LABEL_REPLICA = "torchx.ai/replica"

CONTAINER_STATE_MAP = {
    "created": AppState.SUBMITTED,
    "restarting": AppState.RUNNING,
    "running": AppState.RUNNING,
    "paused": AppState.PENDING,
    "removing": AppState.RUNNING,
    "exited": AppState.SUCCEEDED,  # refined by exit code
    "dead": AppState.FAILED,
}


@dataclass
class DockerContainer:
    image: str
    command: List[str]
    kwargs: Dict[str, Any]


@dataclass
class DockerJob:
    app_id: str
    containers: List[DockerContainer] = field(default_factory=list)

    def __str__(self) -> str:
        return json.dumps(
            {
                "app_id": self.app_id,
                "containers": [
                    {"image": c.image, "command": c.command,
                     "name": c.kwargs.get("name"),
                     "environment": c.kwargs.get("environment"),
                     "devices": c.kwargs.get("devices")}
                    for c in self.containers
                ],
            },
            indent=2,
        )


def _replica_name(app_id: str, role: str, idx: int) -> str:
    return f"{app_id}-{role}-{idx}"


@dataclass
class DockerOpts(StructuredOpts):
    """Typed run options for ``local_docker`` (StructuredOpts parity:
    reference schedulers/api.py:79-324 + docker_scheduler.py:129)."""

    copy_env: Optional[List[str]] = None
    """glob patterns of host env vars to copy into containers"""

    env: Optional[Dict[str, str]] = None
    """extra env vars for all containers"""

    privileged: bool = False
    """run containers privileged"""


class DockerScheduler(DockerWorkspaceMixin, Scheduler[DockerJob]):
    def __init__(self, session_name: str, client: Optional[Any] = None) -> None:
        super().__init__("local_docker", session_name, docker_client=client)
        self.__client = client

    def _client(self) -> Any:
        if self.__client is None:
            import docker

            self.__client = docker.from_env()
        return self.__client

    def run_opts(self) -> runopts:
        return DockerOpts.as_runopts()

    def _ensure_network(self) -> None:
        import docker.errors

        client = self._client()
        # cross-process lock: concurrent creates can BOTH succeed in docker
        # and leave duplicate bridge networks (parity:
        # docker_scheduler.py:105-126 filelock)
        import tempfile

        from filelock import FileLock

        lock_path = os.path.join(tempfile.gettempdir(),
                                 "torchx_amd_docker_network.lock")
        with FileLock(lock_path, timeout=60):
            try:
                client.networks.create(
                    name=NETWORK_NAME, driver="bridge", check_duplicate=True
                )
            except docker.errors.APIError as e:
                if "already exists" not in str(e):
                    raise

    # -- dryrun -------------------------------------------------------------
    def _submit_dryrun(self, app: AppDef,
                       cfg: Mapping[str, Any]) -> AppDryRunInfo[DockerJob]:
        app_id = make_unique(app.name)
        job = DockerJob(app_id=app_id)

        copy_env = cfg.get("copy_env") or []
        extra_env = cfg.get("env") or {}

        rank0_name = None
        for role in app.roles:
            for replica_id in range(role.num_replicas):
                name = _replica_name(app_id, role.name, replica_id)
                if rank0_name is None:
                    rank0_name = name
                values = macros.Values(
                    img_root="",
                    app_id=app_id,
                    replica_id=str(replica_id),
                    rank0_env="TORCHX_RANK0_HOST",
                )
                replica = values.apply(role)
                env = dict(replica.env)
                if copy_env:
                    import fnmatch

                    for pat in copy_env:
                        for k, v in os.environ.items():
                            if fnmatch.fnmatch(k, pat):
                                env.setdefault(k, v)
                env.update(extra_env)
                env["TORCHX_RANK0_HOST"] = rank0_name
                env["TORCHX_IMAGE"] = replica.image
                env.setdefault("PYTHONUNBUFFERED", "1")

                res = replica.resource
                kwargs: Dict[str, Any] = {
                    "name": name,
                    "environment": env,
                    "labels": {
                        LABEL_APP_ID: app_id,
                        LABEL_ROLE: role.name,
                        LABEL_REPLICA: str(replica_id),
                    },
                    "hostname": name,
                    "network": NETWORK_NAME,
                    "detach": True,
                    "privileged": bool(cfg.get("privileged", False)),
                    "devices": [],
                    "mounts": [],
                    "volumes": {},
                }
                if res.cpu > 0:
                    kwargs["nano_cpus"] = int(res.cpu * 1e9)
                if res.memMB > 0:
                    kwargs["mem_limit"] = f"{res.memMB}m"
                    kwargs["shm_size"] = f"{res.memMB}m"
                if res.gpu > 0:
                    # ROCm GPU access: KFD + DRI device nodes; pin the
                    # replica's GPUs via HIP/ROCR_VISIBLE_DEVICES
                    kwargs["devices"] += [
                        "/dev/kfd:/dev/kfd:rwm",
                        "/dev/dri:/dev/dri:rwm",
                    ]
                    kwargs["group_add"] = ["video", "render"]
                    kwargs["security_opt"] = ["seccomp=unconfined"]
                for dm in get_device_mounts(res.devices):
                    kwargs["devices"].append(
                        f"{dm.src_path}:{dm.dst_path}:{dm.permissions}"
                    )
                for m in replica.mounts:
                    if isinstance(m, BindMount):
                        kwargs["volumes"][m.src_path] = {
                            "bind": m.dst_path,
                            "mode": "ro" if m.read_only else "rw",
                        }
                    elif isinstance(m, VolumeMount):
                        kwargs["volumes"][m.src] = {
                            "bind": m.dst_path,
                            "mode": "ro" if m.read_only else "rw",
                        }
                    elif isinstance(m, DeviceMount):
                        kwargs["devices"].append(
                            f"{m.src_path}:{m.dst_path}:{m.permissions}"
                        )
                if role.max_retries > 0:
                    kwargs["restart_policy"] = {
                        "Name": "on-failure",
                        "MaximumRetryCount": role.max_retries,
                    }
                job.containers.append(
                    DockerContainer(
                        image=replica.image,
                        command=[replica.entrypoint, *replica.args],
                        kwargs=kwargs,
                    )
                )
        return AppDryRunInfo(request=job, fmt=str)

    # -- schedule -----------------------------------------------------------
    def schedule(self, dryrun_info: AppDryRunInfo[DockerJob]) -> str:
        client = self._client()
        self._ensure_network()
        req = dryrun_info.request
        import docker.errors

        for c in req.containers:
            try:
                client.images.get(c.image)
            except docker.errors.ImageNotFound:
                client.images.pull(c.image)
        for c in req.containers:
            client.containers.run(c.image, c.command, **c.kwargs)
        return req.app_id

    # -- monitor ------------------------------------------------------------
    def _containers(self, app_id: str) -> List[Any]:
        return self._client().containers.list(
            all=True, filters={"label": f"{LABEL_APP_ID}={app_id}"}
        )

    def describe(self, app_id: str) -> Optional[DescribeAppResponse]:
        containers = self._containers(app_id)
        if not containers:
            return None
        roles: Dict[str, RoleStatus] = {}
        states: List[AppState] = []
        for c in containers:
            role = c.labels.get(LABEL_ROLE, "")
            idx = int(c.labels.get(LABEL_REPLICA, "0"))
            state = CONTAINER_STATE_MAP.get(c.status, AppState.UNKNOWN)
            if c.status == "exited":
                exit_code = c.attrs.get("State", {}).get("ExitCode", 0)
                state = AppState.SUCCEEDED if exit_code == 0 else AppState.FAILED
            states.append(state)
            rs = roles.setdefault(role, RoleStatus(role=role))
            rs.replicas.append(
                ReplicaStatus(id=idx, state=state, role=role,
                              hostname=c.name)
            )
        if any(s == AppState.FAILED for s in states):
            app_state = AppState.FAILED
        elif any(s == AppState.RUNNING for s in states):
            app_state = AppState.RUNNING
        elif all(s == AppState.SUCCEEDED for s in states):
            app_state = AppState.SUCCEEDED
        else:
            app_state = states[0] if states else AppState.UNKNOWN
        return DescribeAppResponse(
            app_id=app_id, state=app_state, roles_statuses=list(roles.values())
        )

    def list(self) -> List[ListAppResponse]:
        containers = self._client().containers.list(
            all=True, filters={"label": LABEL_APP_ID}
        )
        seen = {}
        for c in containers:
            app_id = c.labels[LABEL_APP_ID]
            if app_id not in seen:
                desc = self.describe(app_id)
                seen[app_id] = desc.state if desc else AppState.UNKNOWN
        return [ListAppResponse(app_id=a, state=s) for a, s in seen.items()]

    def _cancel_existing(self, app_id: str) -> None:
        for c in self._containers(app_id):
            try:
                c.stop()
            except Exception:  # noqa: BLE001
                log.warning("failed to stop container %s", c.name)

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
        containers = self._client().containers.list(
            all=True,
            filters={
                "label": [
                    f"{LABEL_APP_ID}={app_id}",
                    f"{LABEL_ROLE}={role_name}",
                    f"{LABEL_REPLICA}={k}",
                ]
            },
        )
        if not containers:
            raise ValueError(f"no container for {app_id}/{role_name}/{k}")
        c = containers[0]
        logs = c.logs(since=since, until=until, stream=should_tail,
                      follow=should_tail)
        if isinstance(logs, bytes):
            lines: Iterable[str] = logs.decode("utf-8",
                                               errors="replace").splitlines()
        else:
            lines = (chunk.decode("utf-8", errors="replace").rstrip("\n")
                     for chunk in logs)
        if regex:
            from .api import filter_regex

            lines = filter_regex(regex, lines)
        return lines


def create_scheduler(session_name: str, **kwargs: Any) -> DockerScheduler:
    return DockerScheduler(session_name=session_name,
                           client=kwargs.get("client"))
