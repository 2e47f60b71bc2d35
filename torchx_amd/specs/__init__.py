"""Public spec API (parity: torchx/specs/__init__.py).

``resource()`` resolves a named resource (``-h mi355x.8gpu``) or builds one
from explicit cpu/gpu/memMB.  Custom named resources come from the
``TORCHX_CUSTOM_NAMED_RESOURCES`` env var (``module:attr`` of a dict) and
registered plugins.
"""

from __future__ import annotations

import importlib
import os
import threading
from typing import Callable, Dict, Optional

from .api import (  # noqa: F401
    AppDef,
    AppDryRunInfo,
    AppHandle,
    AppState,
    AppStatus,
    AppStatusError,
    BindMount,
    ConfigValue,
    DeviceMount,
    InvalidRunConfigException,
    Mount,
    MountType,
    NONE,
    NULL_RESOURCE,
    ReplicaStatus,
    Resource,
    RetryPolicy,
    Role,
    RoleStatus,
    VolumeMount,
    Workspace,
    appdef_to_dict,
    is_terminal,
    macros,
    make_app_handle,
    parse_app_handle,
    replace_role,
    runopt,
    runopts,
)
from .builders import materialize_appdef, parse_mounts  # noqa: F401
from .capabilities import (  # noqa: F401 — public re-exports
    GFX_ARCH,
    HBM_GB,
    XGMI_LINKS,
    CapabilityKey,
)
from .named_resources import NAMED_RESOURCES

_lock = threading.Lock()
_extra_named_resources: Dict[str, Callable[[], Resource]] = {}
_loaded_custom = False


def register_named_resource(name: str, factory: Callable[[], Resource]) -> None:
    with _lock:
        _extra_named_resources[name] = factory


def _load_custom_from_env() -> None:
    global _loaded_custom
    if _loaded_custom:
        return
    _loaded_custom = True
    spec = os.environ.get("TORCHX_CUSTOM_NAMED_RESOURCES")
    if not spec:
        return
    for entry in spec.split(","):
        entry = entry.strip()
        if not entry:
            continue
        mod_name, _, attr = entry.partition(":")
        mod = importlib.import_module(mod_name)
        mapping = getattr(mod, attr or "NAMED_RESOURCES")
        _extra_named_resources.update(mapping)


def named_resources() -> Dict[str, Callable[[], Resource]]:
    from .named_resources_cloud import NAMED_RESOURCES as CLOUD

    _load_custom_from_env()
    out = dict(NAMED_RESOURCES)
    out.update(CLOUD)
    out.update(_extra_named_resources)
    return out


def resource(
    cpu: Optional[int] = None,
    gpu: Optional[int] = None,
    memMB: Optional[int] = None,
    h: Optional[str] = None,
) -> Resource:
    """Named host ``h`` wins over explicit cpu/gpu/memMB (parity:
    torchx/specs/__init__.py:222-255; same defaults cpu=2 gpu=0 memMB=1024)."""
    if h:
        lib = named_resources()
        if h not in lib:
            raise ValueError(
                f"unknown named resource {h!r}; known: {sorted(lib)}"
            )
        return lib[h]()
    return Resource(cpu=cpu or 2, gpu=gpu or 0, memMB=memMB or 1024)


def get_named_resource(h: str) -> Resource:
    return resource(h=h)
