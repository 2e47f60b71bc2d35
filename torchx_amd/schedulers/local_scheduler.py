"""Local scheduler: replicas as subprocess process-groups on this host.

Parity with the reference ``local_cwd`` scheduler
(torchx/schedulers/local_scheduler.py): per-replica stdout/stderr/combined
log files + SUCCESS manifest, macro substitution per replica, two-stage
SIGTERM->SIGKILL teardown, orphan cleanup at interpreter exit, LRU app cache,
log tailing — with the GPU layer MI355X-native: HIP device enumeration via
amd-smi and contiguous ``HIP_VISIBLE_DEVICES``/``ROCR_VISIBLE_DEVICES``
partitioning per replica (devices.py).

Env wire contract kept compatible: TORCHX_RANK0_HOST,
TORCHELASTIC_ERROR_FILE, PET_LOG_DIR, TORCHX_IMAGE, PYTHONUNBUFFERED.
"""

from __future__ import annotations

import atexit
import json
import logging
import os
import shutil
import signal
import subprocess
import tempfile
import time
from dataclasses import dataclass, field
from datetime import datetime
from typing import Any, Dict, Iterable, List, Mapping, Optional

from torchx_amd.specs import (
    AppDef,
    AppDryRunInfo,
    AppState,
    NONE,
    ReplicaStatus,
    RoleStatus,
    is_terminal,
    macros,
    runopts,
)

from .api import (
    DescribeAppResponse, ListAppResponse, Scheduler, Stream, StructuredOpts,
)
from .devices import (
    device_env, hip_device_count, numa_bind_args, partition_devices,
)
from .ids import make_unique
from .streams import Tee

log = logging.getLogger(__name__)

SUCCESS_FILE = "SUCCESS"
TERMINATE_TIMEOUT = 10.0


# ---------------------------------------------------------------------------
# request model
# ---------------------------------------------------------------------------


@dataclass
class ReplicaParam:
    args: List[str]
    env: Dict[str, str]
    stdout: str
    stderr: str
    combined: str
    cwd: Optional[str] = None


@dataclass
class PopenRequest:
    app_id: str
    log_dir: str
    role_params: Dict[str, List[ReplicaParam]] = field(default_factory=dict)

    def __str__(self) -> str:
        view = {
            "app_id": self.app_id,
            "log_dir": self.log_dir,
            "roles": {
                r: [{"args": p.args, "env": {k: v for k, v in p.env.items()
                                             if k.startswith(("TORCHX", "HIP",
                                                              "ROCR", "PET",
                                                              "TORCHELASTIC"))}}
                    for p in ps]
                for r, ps in self.role_params.items()
            },
        }
        return json.dumps(view, indent=2)


# ---------------------------------------------------------------------------
# runtime state
# ---------------------------------------------------------------------------


class _Replica:
    def __init__(self, role: str, idx: int, proc: subprocess.Popen,
                 param: ReplicaParam, tee: Optional[Tee]) -> None:
        self.role = role
        self.idx = idx
        self.proc = proc
        self.param = param
        self.tee = tee
        self.state = AppState.RUNNING

    def poll(self) -> AppState:
        if is_terminal(self.state):
            return self.state
        rc = self.proc.poll()
        if rc is None:
            return AppState.RUNNING
        self.state = AppState.SUCCEEDED if rc == 0 else AppState.FAILED
        if self.tee:
            self.tee.close()
        return self.state

    def terminate(self) -> None:
        if self.proc.poll() is None:
            try:
                os.killpg(self.proc.pid, signal.SIGTERM)
            except ProcessLookupError:
                pass
            deadline = time.time() + TERMINATE_TIMEOUT
            while time.time() < deadline and self.proc.poll() is None:
                time.sleep(0.1)
            if self.proc.poll() is None:
                try:
                    os.killpg(self.proc.pid, signal.SIGKILL)
                except ProcessLookupError:
                    pass
                self.proc.wait()
        if self.tee:
            self.tee.close()
        if not is_terminal(self.state):
            self.state = AppState.CANCELLED


class _LocalApp:
    def __init__(self, app_id: str, log_dir: str) -> None:
        self.app_id = app_id
        self.log_dir = log_dir
        self.replicas: List[_Replica] = []
        self.state = AppState.RUNNING
        self.closed = False

    def poll(self) -> AppState:
        if self.closed:
            return self.state
        states = [r.poll() for r in self.replicas]
        if any(s == AppState.FAILED for s in states):
            # a failed replica fails the app: tear down the rest
            if all(is_terminal(s) for s in states):
                self.close(AppState.FAILED)
            else:
                self.kill(AppState.FAILED)
            return self.state
        if all(s == AppState.SUCCEEDED for s in states):
            self.close(AppState.SUCCEEDED)
            return self.state
        self.state = AppState.RUNNING
        return self.state

    def kill(self, final_state: AppState = AppState.CANCELLED) -> None:
        for r in self.replicas:
            r.terminate()
        self.close(final_state)

    def close(self, final_state: AppState) -> None:
        if self.closed:
            return
        self.state = final_state
        self.closed = True
        try:
            if final_state == AppState.SUCCEEDED:
                with open(os.path.join(self.log_dir, SUCCESS_FILE), "w") as f:
                    json.dump({"app_id": self.app_id, "state": str(final_state)}, f)
        except OSError:
            pass

    def structured_error(self) -> str:
        """Earliest-mtime torchelastic reply file wins (parity:
        local_scheduler.py:418-437)."""
        best: Optional[str] = None
        best_mtime = float("inf")
        for rep in self.replicas:
            ef = rep.param.env.get("TORCHELASTIC_ERROR_FILE", "")
            if ef and os.path.isfile(ef):
                mt = os.path.getmtime(ef)
                if mt < best_mtime:
                    best_mtime = mt
                    best = ef
        if best:
            try:
                with open(best) as f:
                    return f.read()
            except OSError:
                pass
        return NONE


# ---------------------------------------------------------------------------
# scheduler
# ---------------------------------------------------------------------------


_INSTANCES: List["LocalScheduler"] = []


def _cleanup_all() -> None:
    for sched in _INSTANCES:
        sched.close()


atexit.register(_cleanup_all)


@dataclass
class LocalOpts(StructuredOpts):
    """Typed run options for ``local_cwd`` (StructuredOpts parity:
    reference schedulers/api.py:79-324 + local_scheduler.py:177)."""

    log_dir: Optional[str] = None
    """base dir for replica logs (default: tmp dir)"""

    auto_set_hip_visible_devices: bool = True
    """partition host GPUs across replicas via HIP_VISIBLE_DEVICES/ROCR_VISIBLE_DEVICES"""

    prepend_cwd: bool = False
    """put binaries in cwd ahead of PATH (default: PATH wins, cwd appended)"""

    numa_affinity: bool = True
    """wrap each replica in numactl --cpunodebind/--membind for the NUMA node its GPUs hang off (no-op without numactl or a single-node assignment)"""


class LocalScheduler(Scheduler[PopenRequest]):
    """``local_cwd``: runs replica commands from the current working dir."""

    def __init__(self, session_name: str, cache_size: int = 100,
                 extra_paths: Optional[List[str]] = None) -> None:
        super().__init__("local_cwd", session_name)
        self._apps: Dict[str, _LocalApp] = {}
        self._cache_size = cache_size
        _INSTANCES.append(self)

    def run_opts(self) -> runopts:
        return LocalOpts.as_runopts()

    # -- dryrun -------------------------------------------------------------
    def _submit_dryrun(self, app: AppDef,
                       cfg: Mapping[str, Any]) -> AppDryRunInfo[PopenRequest]:
        for role in app.roles:
            if role.mounts:
                raise ValueError(
                    "local_cwd does not support mounts (compat matrix); "
                    "use local_docker or kubernetes"
                )
        app_id = make_unique(app.name)
        base_log = cfg.get("log_dir") or os.path.join(
            tempfile.gettempdir(), "torchx_amd"
        )
        log_dir = os.path.join(base_log, self.session_name, app_id)

        request = PopenRequest(app_id=app_id, log_dir=log_dir)

        device_assignment: Dict[str, List[Optional[str]]] = {}
        if cfg.get("auto_set_hip_visible_devices"):
            device_assignment = partition_devices(
                {r.name: r.num_replicas for r in app.roles},
                {r.name: r.resource.gpu for r in app.roles},
            )

        for role in app.roles:
            params: List[ReplicaParam] = []
            for replica_id in range(role.num_replicas):
                values = macros.Values(
                    img_root=role.image,
                    app_id=app_id,
                    replica_id=str(replica_id),
                    rank0_env="TORCHX_RANK0_HOST",
                )
                replica = values.apply(role)
                replica_log = os.path.join(log_dir, role.name, str(replica_id))
                env = dict(replica.env)
                # PATH precedence (parity: local_scheduler.py:977-990):
                # prepend_cwd puts binaries in cwd ahead of PATH, default
                # appends cwd so PATH binaries win
                base_path = env.get("PATH") or os.environ.get("PATH", "")
                cwd = os.getcwd()
                if cfg.get("prepend_cwd"):
                    env["PATH"] = os.pathsep.join(p for p in (cwd, base_path) if p)
                else:
                    env["PATH"] = os.pathsep.join(p for p in (base_path, cwd) if p)
                env.setdefault("TORCHX_RANK0_HOST", "localhost")
                env.setdefault("TORCHX_IMAGE", role.image)
                env.setdefault("PYTHONUNBUFFERED", "1")
                env.setdefault(
                    "TORCHELASTIC_ERROR_FILE",
                    os.path.join(replica_log, "error.json"),
                )
                env.setdefault("PET_LOG_DIR", replica_log)
                devs = device_assignment.get(role.name)
                numa_prefix: List[str] = []
                if devs is not None:
                    env.update(device_env(devs[replica_id]))
                    if cfg.get("numa_affinity"):
                        numa_prefix = numa_bind_args(devs[replica_id])
                params.append(
                    ReplicaParam(
                        args=[*numa_prefix, replica.entrypoint, *replica.args],
                        env=env,
                        stdout=os.path.join(replica_log, "stdout.log"),
                        stderr=os.path.join(replica_log, "stderr.log"),
                        combined=os.path.join(replica_log, "combined.log"),
                    )
                )
            request.role_params[role.name] = params
        return AppDryRunInfo(request=request, fmt=str)

    def _validate(self, app: AppDef) -> None:
        # local runs tolerate NULL resources (no isolation) — parity with
        # the reference local scheduler.
        pass

    # -- schedule -----------------------------------------------------------
    def schedule(self, dryrun_info: AppDryRunInfo[PopenRequest]) -> str:
        req = dryrun_info.request
        local_app = _LocalApp(req.app_id, req.log_dir)
        os.makedirs(req.log_dir, exist_ok=True)
        try:
            for role_name, params in req.role_params.items():
                for idx, p in enumerate(params):
                    local_app.replicas.append(
                        self._popen(role_name, idx, p)
                    )
        except Exception:
            local_app.kill(AppState.FAILED)
            raise
        self._apps[req.app_id] = local_app
        self._evict_lru()
        return req.app_id

    def _popen(self, role: str, idx: int, p: ReplicaParam) -> _Replica:
        os.makedirs(os.path.dirname(p.stdout), exist_ok=True)
        stdout_f = open(p.stdout, "wb")
        stderr_f = open(p.stderr, "wb")
        combined_f = open(p.combined, "wb")
        env = {**os.environ, **p.env}
        args = list(p.args)
        if not args or not args[0]:
            raise ValueError(f"role {role} replica {idx}: empty entrypoint")
        # bare entrypoint names resolve through the replica PATH (which has
        # cwd appended/prepended per the prepend_cwd opt); explicit relative
        # paths resolve against cwd as Popen does natively
        if os.sep not in args[0] and not os.path.isabs(args[0]):
            resolved = shutil.which(args[0], path=env.get("PATH"))
            if resolved:
                args[0] = resolved
        proc = subprocess.Popen(
            args,
            env=env,
            stdout=stdout_f,
            stderr=stderr_f,
            start_new_session=True,
            cwd=p.cwd,
        )
        # the child owns the fds now; drop the parent-side objects
        stdout_f.close()
        stderr_f.close()
        tee = Tee(combined_f, p.stdout, p.stderr)
        return _Replica(role, idx, proc, p, tee)

    def _evict_lru(self) -> None:
        while len(self._apps) > self._cache_size:
            for app_id, app in self._apps.items():
                if is_terminal(app.poll()):
                    del self._apps[app_id]
                    break
            else:
                oldest = next(iter(self._apps))
                self._apps[oldest].kill()
                del self._apps[oldest]

    # -- monitor ------------------------------------------------------------
    def describe(self, app_id: str) -> Optional[DescribeAppResponse]:
        app = self._apps.get(app_id)
        if app is None:
            return None
        state = app.poll()
        roles: Dict[str, RoleStatus] = {}
        for rep in app.replicas:
            rs = roles.setdefault(rep.role, RoleStatus(role=rep.role))
            rs.replicas.append(
                ReplicaStatus(id=rep.idx, state=rep.state, role=rep.role,
                              hostname="localhost")
            )
        return DescribeAppResponse(
            app_id=app_id,
            state=state,
            structured_error_msg=(
                app.structured_error() if state == AppState.FAILED else NONE
            ),
            roles_statuses=list(roles.values()),
        )

    def list(self) -> List[ListAppResponse]:
        return [
            ListAppResponse(app_id=a, state=app.poll())
            for a, app in self._apps.items()
        ]

    def _cancel_existing(self, app_id: str) -> None:
        app = self._apps.get(app_id)
        if app is not None:
            app.kill()

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
        if since or until:
            log.warning(
                "since/until are ignored by local_cwd log_iter; all log "
                "lines will be returned (parity: local_scheduler.py:1082)"
            )
        app = self._apps.get(app_id)
        if app is None:
            raise ValueError(f"unknown app {app_id}")
        reps = [r for r in app.replicas if r.role == role_name and r.idx == k]
        if not reps:
            raise ValueError(f"no replica {role_name}/{k} in app {app_id}")
        rep = reps[0]
        which = streams or Stream.COMBINED
        path = {
            Stream.STDOUT: rep.param.stdout,
            Stream.STDERR: rep.param.stderr,
            Stream.COMBINED: rep.param.combined,
        }[which]
        it: Iterable[str] = LogIterator(app, path, should_tail)
        if regex:
            from .api import filter_regex

            it = filter_regex(regex, it)
        return it

    def close(self) -> None:
        for app in self._apps.values():
            if not app.closed:
                app.kill()


class LogIterator:
    """Tails a log file until the app finishes (parity:
    local_scheduler.py:1143)."""

    def __init__(self, app: _LocalApp, path: str, should_tail: bool) -> None:
        self._app = app
        self._path = path
        self._tail = should_tail
        self._pos = 0
        self._buf = b""

    def __iter__(self):
        # read in BINARY mode and track byte offsets: seeking a text-mode
        # file by character count corrupts the tail on multibyte UTF-8
        while True:
            try:
                with open(self._path, "rb") as f:
                    f.seek(self._pos)
                    chunk = f.read(65536)
            except OSError:
                chunk = b""
            if chunk:
                self._pos += len(chunk)
                self._buf += chunk
                while True:
                    nl = self._buf.find(b"\n")
                    if nl < 0:
                        break
                    yield self._buf[:nl].decode("utf-8", errors="replace")
                    self._buf = self._buf[nl + 1:]
            else:
                finished = is_terminal(self._app.poll())
                if finished or not self._tail:
                    if self._buf:
                        yield self._buf.decode("utf-8", errors="replace")
                    return
                time.sleep(0.1)


def create_scheduler(session_name: str, **kwargs: Any) -> LocalScheduler:
    return LocalScheduler(session_name=session_name, **kwargs)
