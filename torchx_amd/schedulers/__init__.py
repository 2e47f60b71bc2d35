"""Scheduler registry: deferred-import factory map (parity:
torchx/schedulers/__init__.py).  Plugins may replace the map wholesale via
the ``torchx_amd.schedulers`` entry-point group; the first entry is the
default scheduler."""

from __future__ import annotations

import importlib
from typing import Any, Callable, Dict

from .api import Scheduler  # noqa: F401

DEFAULT_SCHEDULER_MODULES: Dict[str, str] = {
    "local_cwd": "torchx_amd.schedulers.local_scheduler",
    "local_docker": "torchx_amd.schedulers.docker_scheduler",
    "slurm": "torchx_amd.schedulers.slurm_scheduler",
    "kubernetes": "torchx_amd.schedulers.kubernetes_scheduler",
}

SchedulerFactory = Callable[..., Scheduler]


def _deferred(module: str) -> SchedulerFactory:
    def factory(session_name: str, **kwargs: Any) -> Scheduler:
        mod = importlib.import_module(module)
        return mod.create_scheduler(session_name=session_name, **kwargs)

    return factory


def get_scheduler_factories() -> Dict[str, SchedulerFactory]:
    factories = {
        name: _deferred(mod) for name, mod in DEFAULT_SCHEDULER_MODULES.items()
    }
    try:
        from torchx_amd.plugins import registry

        # merge over the defaults (plugins add or override by name; they
        # never silently remove the builtins)
        factories.update(registry().scheduler_factories())
    except Exception:  # noqa: BLE001 — plugins must never break core
        pass
    return factories


def get_default_scheduler_name() -> str:
    return next(iter(get_scheduler_factories()))
