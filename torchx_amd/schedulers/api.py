"""Scheduler abstraction: dryrun-first submit protocol.

Parity with the reference Scheduler ABC (torchx/schedulers/api.py:373):
``submit = submit_dryrun + schedule``; ``describe/cancel/delete/list/
log_iter`` monitor the data plane.  Run-config schemas are ``runopts``
(torchx_amd.specs.api).
"""

from __future__ import annotations

import abc
import re
from dataclasses import dataclass, field
from datetime import datetime
from enum import Enum
from typing import Any, Dict, Generic, Iterable, List, Mapping, Optional, TypeVar

from torchx_amd.specs import (
    AppDef,
    AppDryRunInfo,
    AppState,
    NULL_RESOURCE,
    ReplicaStatus,
    RoleStatus,
    Role,
    runopts,
)

T = TypeVar("T")


class Stream(str, Enum):
    STDOUT = "stdout"
    STDERR = "stderr"
    COMBINED = "combined"


@dataclass
class DescribeAppResponse:
    app_id: str = "<NOT_SET>"
    state: AppState = AppState.UNSUBMITTED
    num_restarts: int = -1
    msg: str = ""
    structured_error_msg: str = "<NONE>"
    ui_url: Optional[str] = None
    roles_statuses: List[RoleStatus] = field(default_factory=list)
    roles: List[Role] = field(default_factory=list)


@dataclass
class ListAppResponse:
    app_id: str
    state: AppState
    app_handle: str = "<NOT_SET>"
    name: str = ""


class Scheduler(abc.ABC, Generic[T]):
    """Backend adapter.  Subclasses implement ``_submit_dryrun`` producing a
    scheduler-native request and ``schedule`` consuming it."""

    def __init__(self, backend: str, session_name: str) -> None:
        self.backend = backend
        self.session_name = session_name

    # -- submit -------------------------------------------------------------
    def submit(self, app: AppDef, cfg: Mapping[str, Any],
               workspace: Optional[str] = None) -> str:
        dryrun_info = self.submit_dryrun(app, cfg)
        return self.schedule(dryrun_info)

    def submit_dryrun(self, app: AppDef, cfg: Mapping[str, Any]) -> AppDryRunInfo[T]:
        resolved = self.run_opts().resolve(cfg)
        self._validate(app)
        info = self._submit_dryrun(app, resolved)
        for role in app.roles:
            info = role.pre_proc(self.backend, info)
        info._app = app
        info._cfg = dict(resolved)
        info._scheduler = self.backend
        return info

    @abc.abstractmethod
    def _submit_dryrun(self, app: AppDef, cfg: Mapping[str, Any]) -> AppDryRunInfo[T]:
        ...

    @abc.abstractmethod
    def schedule(self, dryrun_info: AppDryRunInfo[T]) -> str:
        ...

    # -- monitoring ---------------------------------------------------------
    @abc.abstractmethod
    def describe(self, app_id: str) -> Optional[DescribeAppResponse]:
        ...

    def exists(self, app_id: str) -> bool:
        return self.describe(app_id) is not None

    @abc.abstractmethod
    def _cancel_existing(self, app_id: str) -> None:
        ...

    def cancel(self, app_id: str) -> None:
        if self.exists(app_id):
            self._cancel_existing(app_id)

    def delete(self, app_id: str) -> None:
        self.cancel(app_id)

    def list(self) -> List[ListAppResponse]:
        raise NotImplementedError(
            f"{self.backend} scheduler does not support listing apps"
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
        raise NotImplementedError(
            f"{self.backend} scheduler does not support log iteration"
        )

    # -- config -------------------------------------------------------------
    def run_opts(self) -> runopts:
        return runopts()

    def close(self) -> None:
        pass

    # -- validation ---------------------------------------------------------
    def _pre_build_validate(self, app: AppDef, cfg: Mapping[str, Any]) -> None:
        pass

    def _validate(self, app: AppDef) -> None:
        for role in app.roles:
            if role.resource is NULL_RESOURCE:
                raise ValueError(
                    f"role {role.name!r} has no resource; use specs.resource() "
                    "or a named resource (e.g. -h mi355x.8gpu)"
                )


# -- log helpers (parity: schedulers/api.py:541-567) -------------------------


def filter_regex(regex: str, data: Iterable[str]) -> Iterable[str]:
    r = re.compile(regex)
    return filter(lambda x: r.search(x), data)


def split_lines(text: str) -> List[str]:
    """Split keeping trailing newlines, preserving a partial last line."""
    return text.splitlines(keepends=True)


def split_lines_iterator(chunks: Iterable[str]) -> Iterable[str]:
    buf = ""
    for chunk in chunks:
        buf += chunk
        while True:
            idx = buf.find("\n")
            if idx < 0:
                break
            yield buf[: idx + 1]
            buf = buf[idx + 1:]
    if buf:
        yield buf
