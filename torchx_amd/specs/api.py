"""Job specification data model for the MI355X-native launcher.

This is the layer-1 data model: an ``AppDef`` describes a distributed
application as a list of ``Role``s (homogeneous gangs of replicas), each with a
``Resource`` shape sized for MI355X nodes (288 GB HBM3E per GPU).  Schedulers
consume AppDefs and produce scheduler-native requests.

Behavioral parity with the reference spec layer (torchx/specs/api.py: Resource
:105, macros :198, RetryPolicy :303, mounts :339-363, Role :573, AppDef :683,
AppState :704, AppStatus :771, AppDryRunInfo :999, runopts :1124, handles
:1458), re-designed from scratch.
"""

from __future__ import annotations

import json
import os
import re
from dataclasses import dataclass, field, asdict
from datetime import datetime
from enum import Enum
from string import Template
from typing import Any, Callable, Dict, Generic, Iterator, List, Mapping, Optional, Tuple, TypeVar, Union

# ---------------------------------------------------------------------------
# Resource
# ---------------------------------------------------------------------------


@dataclass
class Resource:
    """Compute shape of one replica of a role.

    ``gpu`` counts MI355X devices; ``memMB`` is host memory.  Device memory is
    implied (288 GB HBM3E per GPU) and may be recorded in ``capabilities`` under
    the key ``amd.com/hbm_gb``.
    """

    cpu: int
    gpu: int
    memMB: int
    capabilities: Dict[str, Any] = field(default_factory=dict)
    devices: Dict[str, int] = field(default_factory=dict)

    @staticmethod
    def copy(original: "Resource", **capabilities: Any) -> "Resource":
        res = Resource(
            cpu=original.cpu,
            gpu=original.gpu,
            memMB=original.memMB,
            capabilities={**original.capabilities},
            devices={**original.devices},
        )
        res.capabilities.update(capabilities)
        return res


# Sentinels: a NULL resource means "scheduler must fill it in" (mirrors
# reference NULL_RESOURCE, torchx/specs/api.py:173).
NULL_RESOURCE: Resource = Resource(cpu=-1, gpu=-1, memMB=-1)

# Used for schedulers that take resources out-of-band (e.g. named queues).
RESOURCE_OPAQUE = NULL_RESOURCE


# ---------------------------------------------------------------------------
# Macros
# ---------------------------------------------------------------------------


class macros:
    """Substitution macros usable in Role args/env; resolved per replica at
    submit time (reference: torchx/specs/api.py:198-300).

    ::

        img_root    - image root directory
        app_id      - scheduler-assigned application id
        replica_id  - index of the replica within the role
        rank0_env   - name of the env var holding the rank-0 host
    """

    img_root = "${img_root}"
    app_id = "${app_id}"
    replica_id = "${replica_id}"
    rank0_env = "${rank0_env}"

    @dataclass
    class Values:
        img_root: str
        app_id: str
        replica_id: str
        rank0_env: str

        def apply(self, role: "Role") -> "Role":
            """Return a deep-substituted copy of ``role``."""
            sub = self.substitute
            replica = replace_role(
                role,
                args=[sub(a) for a in role.args],
                env={k: sub(v) for k, v in role.env.items()},
                metadata=_substitute_obj(role.metadata, sub),
            )
            return replica

        def substitute(self, arg: str) -> str:
            return Template(arg).safe_substitute(
                img_root=self.img_root,
                app_id=self.app_id,
                replica_id=self.replica_id,
                rank0_env=self.rank0_env,
            )


def _substitute_obj(obj: Any, sub: Callable[[str], str]) -> Any:
    if isinstance(obj, str):
        return sub(obj)
    if isinstance(obj, dict):
        return {k: _substitute_obj(v, sub) for k, v in obj.items()}
    if isinstance(obj, list):
        return [_substitute_obj(v, sub) for v in obj]
    return obj


# ---------------------------------------------------------------------------
# Retry / mounts / workspace
# ---------------------------------------------------------------------------


class RetryPolicy(str, Enum):
    """What to restart when a replica fails (reference api.py:303-330)."""

    REPLICA = "REPLICA"
    APPLICATION = "APPLICATION"
    ROLE = "ROLE"

    def __str__(self) -> str:  # keep str(RetryPolicy.REPLICA) == "REPLICA"
        return self.value


class MountType(str, Enum):
    BIND = "bind"
    VOLUME = "volume"
    DEVICE = "device"


@dataclass
class BindMount:
    src_path: str
    dst_path: str
    read_only: bool = False


@dataclass
class VolumeMount:
    src: str
    dst_path: str
    read_only: bool = False


@dataclass
class DeviceMount:
    src_path: str
    dst_path: str
    permissions: str = "rwm"


Mount = Union[BindMount, VolumeMount, DeviceMount]


@dataclass
class Workspace:
    """Local project directories to overlay onto the role image at build
    time. ``projects`` maps a local dir -> destination path inside the
    image root ("" = image root). Multiple projects are merged (later
    entries win on conflicts) before the build (parity:
    torchx/workspace/api.py:97-179)."""

    projects: Dict[str, str] = field(default_factory=dict)

    @staticmethod
    def from_str(workspace: Optional[str]) -> "Workspace":
        """``"dir"`` -> single unmapped project; ``"a:dst1,b:dst2"`` ->
        mapped multi-project."""
        if not workspace:
            return Workspace()
        if isinstance(workspace, Workspace):
            return workspace
        projects: Dict[str, str] = {}
        for part in str(workspace).split(","):
            part = part.strip()
            if not part:
                continue
            src, _, dst = part.partition(":")
            projects[src] = dst
        return Workspace(projects=projects)

    def is_unmapped_single_project(self) -> bool:
        return (len(self.projects) == 1
                and not next(iter(self.projects.values())))

    def __bool__(self) -> bool:
        return bool(self.projects)


# ---------------------------------------------------------------------------
# Role / AppDef
# ---------------------------------------------------------------------------


@dataclass
class Role:
    """A homogeneous gang of ``num_replicas`` processes/containers.

    ``entrypoint`` + ``args`` run inside ``image`` with ``env``.  ``resource``
    sizes one replica.  ``port_map`` declares named ports (e.g. the c10d
    rendezvous port).  ``min_replicas`` enables elastic gangs.
    """

    name: str
    image: str
    entrypoint: str = ""
    args: List[str] = field(default_factory=list)
    env: Dict[str, str] = field(default_factory=dict)
    num_replicas: int = 1
    min_replicas: Optional[int] = None
    max_retries: int = 0
    retry_policy: RetryPolicy = RetryPolicy.APPLICATION
    resource: Resource = field(default_factory=lambda: NULL_RESOURCE)
    port_map: Dict[str, int] = field(default_factory=dict)
    metadata: Dict[str, Any] = field(default_factory=dict)
    mounts: List[Mount] = field(default_factory=list)

    def pre_proc(self, scheduler: str, dryrun_info: "AppDryRunInfo") -> "AppDryRunInfo":
        """Scheduler-specific hook applied during submit_dryrun."""
        return dryrun_info


def replace_role(role: Role, **overrides: Any) -> Role:
    kwargs: Dict[str, Any] = dict(
        name=role.name,
        image=role.image,
        entrypoint=role.entrypoint,
        args=list(role.args),
        env=dict(role.env),
        num_replicas=role.num_replicas,
        min_replicas=role.min_replicas,
        max_retries=role.max_retries,
        retry_policy=role.retry_policy,
        resource=role.resource,
        port_map=dict(role.port_map),
        metadata=dict(role.metadata),
        mounts=list(role.mounts),
    )
    kwargs.update(overrides)
    return Role(**kwargs)


@dataclass
class AppDef:
    """A distributed application: a named list of Roles."""

    name: str
    roles: List[Role] = field(default_factory=list)
    metadata: Dict[str, str] = field(default_factory=dict)

    def __str__(self) -> str:
        return json.dumps(appdef_to_dict(self), indent=2)


def appdef_to_dict(app: AppDef) -> Dict[str, Any]:
    d = asdict(app)
    for role in d["roles"]:
        role["retry_policy"] = str(role["retry_policy"])
    return d


# ---------------------------------------------------------------------------
# App status
# ---------------------------------------------------------------------------


class AppState(int, Enum):
    """Lifecycle states (reference api.py:704-770); terminal states are
    SUCCEEDED/FAILED/CANCELLED."""

    UNSUBMITTED = 0
    SUBMITTED = 1
    PENDING = 2
    RUNNING = 3
    SUCCEEDED = 4
    FAILED = 5
    CANCELLED = 6
    UNKNOWN = 7

    def __str__(self) -> str:
        return self.name


_TERMINAL_STATES = frozenset(
    {AppState.SUCCEEDED, AppState.FAILED, AppState.CANCELLED}
)


def is_terminal(state: AppState) -> bool:
    return state in _TERMINAL_STATES


# Ordered so max() picks the "most representative" role state.
NONE: str = "<NONE>"


@dataclass
class ReplicaStatus:
    id: int
    state: AppState
    role: str
    hostname: str = ""
    structured_error_msg: str = NONE


@dataclass
class RoleStatus:
    role: str
    replicas: List[ReplicaStatus] = field(default_factory=list)


@dataclass
class AppStatus:
    """Status of a submitted app, with structured error extraction from
    torchelastic-style reply files (reference api.py:871-930)."""

    state: AppState
    num_restarts: int = 0
    msg: str = ""
    structured_error_msg: str = NONE
    ui_url: Optional[str] = None
    roles: List[RoleStatus] = field(default_factory=list)

    def is_terminal(self) -> bool:
        return is_terminal(self.state)

    def raise_for_status(self) -> None:
        if self.state != AppState.SUCCEEDED:
            raise AppStatusError(self, f"job did not succeed: {self}")

    def _error_message(self) -> Optional[Dict[str, Any]]:
        if self.structured_error_msg == NONE:
            return None
        try:
            body = json.loads(self.structured_error_msg)
        except json.JSONDecodeError:
            return {"message": self.structured_error_msg}
        # torchelastic reply-file schema: {"message": {"message": ..,
        # "extraInfo": {"py_callstack": ..., "timestamp": ...}}}
        msg = body.get("message", body)
        if isinstance(msg, str):
            return {"message": msg}
        return msg

    def format(self) -> str:
        lines = [f"AppStatus:", f"  state: {self.state}", f"  num_restarts: {self.num_restarts}"]
        if self.msg:
            lines.append(f"  msg: {self.msg}")
        if self.ui_url:
            lines.append(f"  ui_url: {self.ui_url}")
        err = self._error_message()
        if err:
            message = err.get("message", "")
            extra = err.get("extraInfo", {})
            ts = extra.get("timestamp")
            if ts:
                try:
                    message += f" (at {datetime.fromtimestamp(int(ts))})"
                except (ValueError, OverflowError):
                    pass
            lines.append(f"  error: {message}")
            stack = extra.get("py_callstack")
            if stack:
                lines.append("  callstack:")
                lines.extend("    " + ln for ln in str(stack).splitlines())
        for rs in self.roles:
            for rep in rs.replicas:
                lines.append(
                    f"  {rs.role}[{rep.id}]: {rep.state}"
                    + (f" ({rep.hostname})" if rep.hostname else "")
                )
        return "\n".join(lines)

    def __str__(self) -> str:
        return f"AppStatus(state={self.state}, num_restarts={self.num_restarts}, msg={self.msg!r})"


class AppStatusError(Exception):
    def __init__(self, status: AppStatus, message: str) -> None:
        super().__init__(message)
        self.status = status


# ---------------------------------------------------------------------------
# Dryrun info / handles
# ---------------------------------------------------------------------------

T = TypeVar("T")


@dataclass
class AppDryRunInfo(Generic[T]):
    """The scheduler-native request produced by ``submit_dryrun`` plus a
    printable form (reference api.py:999)."""

    request: T
    fmt: Callable[[T], str]
    _app: Optional[AppDef] = None
    _cfg: Optional[Dict[str, Any]] = None
    _scheduler: Optional[str] = None

    def __str__(self) -> str:
        return self.fmt(self.request)


AppHandle = str
_APP_HANDLE_RE = re.compile(
    r"^(?P<scheduler>[\w\-+.]+)://(?P<session>[\w\-+.@=]*)/(?P<app_id>.+)$"
)


def make_app_handle(scheduler_backend: str, session_name: str, app_id: str) -> AppHandle:
    return f"{scheduler_backend}://{session_name}/{app_id}"


def parse_app_handle(app_handle: AppHandle) -> Tuple[str, str, str]:
    """``scheduler://session/app_id`` -> (scheduler, session, app_id)."""
    m = _APP_HANDLE_RE.match(app_handle)
    if not m:
        raise ValueError(
            f"malformed app handle {app_handle!r}; expected scheduler://session/app_id"
        )
    return m.group("scheduler"), m.group("session"), m.group("app_id")


# ---------------------------------------------------------------------------
# runopts — per-scheduler typed run config
# ---------------------------------------------------------------------------

ConfigValue = Union[str, int, float, bool, List[str], Dict[str, str], None]


class InvalidRunConfigException(Exception):
    pass


@dataclass
class runopt:
    name: str
    opt_type: type
    default: ConfigValue
    required: bool
    help: str

    def cast(self, value: ConfigValue) -> ConfigValue:
        if value is None:
            return None
        t = self.opt_type
        if t is bool:
            if isinstance(value, bool):
                return value
            return str(value).strip().lower() in ("1", "true", "yes", "on")
        if t in (int, float, str):
            return t(value)  # type: ignore[call-arg]
        if t is List[str] or t is list:
            if isinstance(value, list):
                return value
            return [p for p in re.split(r"[,;]", str(value)) if p]
        if t is Dict[str, str] or t is dict:
            if isinstance(value, dict):
                return value
            out: Dict[str, str] = {}
            for pair in re.split(r"[,;]", str(value)):
                if not pair:
                    continue
                k, _, v = pair.partition(":")
                if not _:
                    k, _, v = pair.partition("=")
                out[k.strip()] = v.strip()
            return out
        return value


class runopts:
    """Typed run-config schema for a scheduler: declare with :py:meth:`add`,
    parse strings with :py:meth:`cfg_from_str`, validate with
    :py:meth:`resolve` (reference api.py:1124-1325)."""

    def __init__(self) -> None:
        self._opts: Dict[str, runopt] = {}

    def add(
        self,
        cfg_key: str,
        type_: type,
        help: str,
        default: ConfigValue = None,
        required: bool = False,
    ) -> None:
        self._opts[cfg_key] = runopt(cfg_key, type_, default, required, help)

    def get(self, name: str) -> Optional[runopt]:
        return self._opts.get(name)

    def update(self, other: "runopts") -> None:
        self._opts.update(other._opts)

    def __iter__(self) -> Iterator[Tuple[str, runopt]]:
        return iter(self._opts.items())

    def cfg_from_str(self, cfg_str: str) -> Dict[str, ConfigValue]:
        """Parse ``"k1=v1,k2=v2;k3=v3"``; list values use ``,`` within a
        ``k=v1,v2`` group (split happens on the LAST ``=``-free segments)."""
        cfg: Dict[str, ConfigValue] = {}
        if not cfg_str:
            return cfg
        # split on , and ; but re-join segments with no '=' into the
        # previous key's list value
        last_key: Optional[str] = None
        for token in re.split(r"[,;]", cfg_str.strip()):
            if not token:
                continue
            if "=" in token:
                k, _, v = token.partition("=")
                k = k.strip()
                opt = self.get(k)
                if opt is not None:
                    cfg[k] = opt.cast(v)
                else:
                    cfg[k] = v
                last_key = k
            elif last_key is not None:
                prev = cfg[last_key]
                if isinstance(prev, dict) and ":" in token:
                    dk, _, dv = token.partition(":")
                    prev[dk.strip()] = dv.strip()
                elif isinstance(prev, list):
                    prev.append(token)
                else:
                    cfg[last_key] = [str(prev), token]
        return cfg

    def resolve(self, cfg: Mapping[str, ConfigValue]) -> Dict[str, ConfigValue]:
        resolved: Dict[str, ConfigValue] = {}
        for name, opt in self._opts.items():
            if name in cfg and cfg[name] is not None:
                resolved[name] = opt.cast(cfg[name])
            elif opt.default is not None:
                resolved[name] = opt.default
            elif opt.required:
                raise InvalidRunConfigException(
                    f"required run option {name!r} not provided; known cfg: {dict(cfg)}"
                )
            else:
                resolved[name] = None
        # pass through unknown keys (schedulers may accept extras)
        for k, v in cfg.items():
            if k not in resolved:
                resolved[k] = v
        return resolved

    def __str__(self) -> str:
        rows = []
        for name, opt in self._opts.items():
            t = getattr(opt.opt_type, "__name__", str(opt.opt_type))
            req = "required" if opt.required else f"default: {opt.default}"
            rows.append(f"    {name} ({t}, {req}): {opt.help}")
        return "runopts:\n" + "\n".join(rows)


# ---------------------------------------------------------------------------
# Error-file helpers (torchelastic wire contract)
# ---------------------------------------------------------------------------


def read_structured_error(error_file: str) -> Optional[str]:
    """Read a torchelastic-style JSON reply file if present."""
    if error_file and os.path.isfile(error_file):
        try:
            with open(error_file) as f:
                return f.read()
        except OSError:
            return None
    return None


def get_type_name(tp: type) -> str:
    return getattr(tp, "__name__", str(tp))
