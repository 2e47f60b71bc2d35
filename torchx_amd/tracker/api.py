"""Experiment tracker API (parity: torchx/tracker/api.py: TrackerBase :56,
AppRun :188 with run_from_env :229).

The launcher injects TORCHX_JOB_ID / TORCHX_TRACKERS /
TORCHX_TRACKER_<NAME>_CONFIG / TORCHX_PARENT_RUN_ID into every role's env
(runner/api.py); inside the job, ``AppRun.run_from_env()`` builds the
configured tracker backends and fans writes out to all of them."""

from __future__ import annotations

import abc
import importlib
import logging
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Iterable, List, Mapping, Optional

log = logging.getLogger(__name__)

ENV_TORCHX_JOB_ID = "TORCHX_JOB_ID"
ENV_TORCHX_TRACKERS = "TORCHX_TRACKERS"
ENV_TORCHX_PARENT_RUN_ID = "TORCHX_PARENT_RUN_ID"


class TrackerBase(abc.ABC):
    """Backend interface: artifacts, metadata, lineage."""

    @abc.abstractmethod
    def add_artifact(self, run_id: str, name: str, path: str,
                     metadata: Optional[Mapping[str, object]] = None) -> None:
        ...

    @abc.abstractmethod
    def artifacts(self, run_id: str) -> Mapping[str, str]:
        ...

    @abc.abstractmethod
    def add_metadata(self, run_id: str, **kwargs: object) -> None:
        ...

    @abc.abstractmethod
    def metadata(self, run_id: str) -> Mapping[str, object]:
        ...

    @abc.abstractmethod
    def add_source(self, run_id: str, source_id: str,
                   artifact_name: Optional[str] = None) -> None:
        ...

    @abc.abstractmethod
    def sources(self, run_id: str,
                artifact_name: Optional[str] = None) -> Iterable[str]:
        ...

    @abc.abstractmethod
    def run_ids(self, **kwargs: str) -> Iterable[str]:
        ...

    def lineage(self, run_id: str) -> Iterable[str]:
        return self.sources(run_id)


def _build_tracker(name: str, config: Optional[str]) -> Optional[TrackerBase]:
    """name is either a module path with a ``create(config)`` factory or a
    plugin-registered tracker name."""
    try:
        from torchx_amd.plugins import registry

        factory = registry().tracker_factory(name)
        if factory is not None:
            return factory(config)
    except Exception:  # noqa: BLE001
        pass
    candidates = [name]
    if name == "fsspec":
        candidates = ["torchx_amd.tracker.fsspec"]
    for cand in candidates:
        try:
            mod = importlib.import_module(cand)
            create = getattr(mod, "create", None)
            if create:
                return create(config)
        except ImportError:
            continue
    log.warning("could not build tracker %r", name)
    return None


def trackers_from_environ(
    env: Optional[Mapping[str, str]] = None,
) -> List[TrackerBase]:
    env = env if env is not None else os.environ
    names = [n for n in env.get(ENV_TORCHX_TRACKERS, "").split(",") if n]
    out: List[TrackerBase] = []
    for name in names:
        cfg = env.get(f"TORCHX_TRACKER_{name.upper()}_CONFIG")
        t = _build_tracker(name, cfg)
        if t is not None:
            out.append(t)
    return out


@dataclass
class AppRun:
    """Fan-out facade used inside training apps."""

    id: str
    backends: List[TrackerBase] = field(default_factory=list)

    _run: Optional["AppRun"] = None

    @classmethod
    def run_from_env(cls, env: Optional[Mapping[str, str]] = None) -> "AppRun":
        if cls._run is not None and env is None:
            return cls._run
        env_map = env if env is not None else os.environ
        run_id = env_map.get(ENV_TORCHX_JOB_ID, "<unset_run_id>")
        run = cls(id=run_id, backends=trackers_from_environ(env_map))
        parent = env_map.get(ENV_TORCHX_PARENT_RUN_ID)
        if parent:
            for b in run.backends:
                try:
                    b.add_source(run_id, parent)
                except Exception:  # noqa: BLE001
                    log.exception("add_source failed")
        if env is None:
            cls._run = run
        return run

    def add_metadata(self, **kwargs: object) -> None:
        for b in self.backends:
            b.add_metadata(self.id, **kwargs)

    def add_artifact(self, name: str, path: str,
                     metadata: Optional[Mapping[str, object]] = None) -> None:
        for b in self.backends:
            b.add_artifact(self.id, name, path, metadata)

    def add_source(self, source_id: str,
                   artifact_name: Optional[str] = None) -> None:
        for b in self.backends:
            b.add_source(self.id, source_id, artifact_name)


def app_run_from_env() -> AppRun:
    return AppRun.run_from_env()
