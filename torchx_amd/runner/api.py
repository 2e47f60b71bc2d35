"""Runner: the session facade over all schedulers (parity:
torchx/runner/api.py:93-709).

run_component -> component resolution -> materialize -> dryrun (deepcopy,
validate, tracker/session env injection, workspace build) -> schedule;
plus status/wait/cancel/delete/describe/log_lines/list, each wrapped in
telemetry events."""

from __future__ import annotations

import copy
import json
import logging
import os
import time
from datetime import datetime
from types import TracebackType
from typing import Any, Dict, Iterable, List, Mapping, Optional, Type

from torchx_amd.schedulers import (
    Scheduler,
    SchedulerFactory,
    get_scheduler_factories,
)
from torchx_amd.schedulers.api import DescribeAppResponse, ListAppResponse, Stream
from torchx_amd.specs import (
    AppDef,
    AppDryRunInfo,
    AppHandle,
    AppState,
    AppStatus,
    ConfigValue,
    is_terminal,
    make_app_handle,
    materialize_appdef,
    parse_app_handle,
)
from torchx_amd.specs.finder import get_component

from . import config as torchx_config
from .events import log_event

log = logging.getLogger(__name__)

NONE = "<NONE>"


def _session_id() -> str:
    from torchx_amd.schedulers.ids import random_id

    return random_id(12)


class Runner:
    def __init__(
        self,
        name: str,
        scheduler_factories: Optional[Dict[str, SchedulerFactory]] = None,
        component_defaults: Optional[Dict[str, Dict[str, str]]] = None,
    ) -> None:
        self._name = name
        self._factories = scheduler_factories or get_scheduler_factories()
        self._schedulers: Dict[str, Scheduler] = {}
        self._component_defaults = component_defaults or {}
        self._session_id = _session_id()

    # -- context ------------------------------------------------------------
    def __enter__(self) -> "Runner":
        return self

    def __exit__(self, etype: Optional[Type[BaseException]],
                 e: Optional[BaseException],
                 tb: Optional[TracebackType]) -> None:
        self.close()

    def close(self) -> None:
        for sched in self._schedulers.values():
            sched.close()

    # -- scheduler access ---------------------------------------------------
    def _scheduler(self, scheduler: str) -> Scheduler:
        if scheduler not in self._schedulers:
            factory = self._factories.get(scheduler)
            if factory is None:
                raise KeyError(
                    f"unknown scheduler {scheduler!r}; "
                    f"registered: {sorted(self._factories)}"
                )
            # TORCHX_<SCHED>_<PARAM> env passthrough (reference api.py:131)
            params: Dict[str, str] = {}
            prefix = f"TORCHX_{scheduler.upper()}_"
            for k, v in os.environ.items():
                if k.startswith(prefix):
                    params[k[len(prefix):].lower()] = v
            self._schedulers[scheduler] = factory(
                session_name=self._name, **params
            )
        return self._schedulers[scheduler]

    def scheduler_backends(self) -> List[str]:
        return list(self._factories)

    def scheduler_run_opts(self, scheduler: str):
        return self._scheduler(scheduler).run_opts()

    # -- submit path --------------------------------------------------------
    def run_component(
        self,
        component: str,
        component_args: List[str],
        scheduler: str = "local_cwd",
        cfg: Optional[Mapping[str, ConfigValue]] = None,
        workspace: Optional[str] = None,
        parent_run_id: Optional[str] = None,
    ) -> AppHandle:
        dryrun_info = self.dryrun_component(
            component, component_args, scheduler, cfg=cfg,
            workspace=workspace, parent_run_id=parent_run_id,
        )
        return self.schedule(dryrun_info)

    def dryrun_component(
        self,
        component: str,
        component_args: List[str],
        scheduler: str = "local_cwd",
        cfg: Optional[Mapping[str, ConfigValue]] = None,
        workspace: Optional[str] = None,
        parent_run_id: Optional[str] = None,
    ) -> AppDryRunInfo:
        comp = get_component(component)
        defaults = self._component_defaults.get(component)
        app = materialize_appdef(comp.fn, component_args, defaults=defaults)
        return self.dryrun(app, scheduler, cfg=cfg, workspace=workspace,
                           parent_run_id=parent_run_id)

    def run(
        self,
        app: AppDef,
        scheduler: str = "local_cwd",
        cfg: Optional[Mapping[str, ConfigValue]] = None,
        workspace: Optional[str] = None,
        parent_run_id: Optional[str] = None,
    ) -> AppHandle:
        dryrun_info = self.dryrun(app, scheduler, cfg=cfg,
                                  workspace=workspace,
                                  parent_run_id=parent_run_id)
        return self.schedule(dryrun_info)

    def dryrun(
        self,
        app: AppDef,
        scheduler: str = "local_cwd",
        cfg: Optional[Mapping[str, ConfigValue]] = None,
        workspace: Optional[str] = None,
        parent_run_id: Optional[str] = None,
    ) -> AppDryRunInfo:
        # deepcopy so the caller's AppDef is never mutated (reference :447)
        app = copy.deepcopy(app)
        cfg = dict(cfg or {})
        torchx_config.load(scheduler, cfg)

        if not app.roles:
            raise ValueError("AppDef has no roles")
        for role in app.roles:
            if not role.entrypoint:
                raise ValueError(f"role {role.name} has no entrypoint")
            if role.num_replicas <= 0:
                raise ValueError(f"role {role.name} has num_replicas<=0")

        sched = self._scheduler(scheduler)
        with log_event("dryrun", scheduler, self._name,
                       runcfg=json.dumps({k: str(v) for k, v in cfg.items()}),
                       workspace=workspace):
            sched._pre_build_validate(app, cfg)

            # tracker/session env injection (reference api.py:400-424);
            # parent run id falls back to the launcher's own env so lineage
            # chains when a tracked job launches further jobs
            parent_run_id = parent_run_id or os.environ.get(
                "TORCHX_PARENT_RUN_ID")
            trackers = torchx_config.get_configured_trackers()
            for role in app.roles:
                role.env.setdefault("TORCHX_INTERNAL_SESSION_ID",
                                    self._session_id)
                if parent_run_id:
                    role.env.setdefault("TORCHX_PARENT_RUN_ID", parent_run_id)
                if trackers:
                    role.env.setdefault("TORCHX_TRACKERS",
                                        ",".join(trackers.keys()))
                    for tname, tcfg in trackers.items():
                        if tcfg:
                            key = f"TORCHX_TRACKER_{tname.upper()}_CONFIG"
                            role.env.setdefault(key, tcfg)

            # TORCHX_JOB_ID must be set BEFORE submit_dryrun materializes the
            # request (reference api.py:400-459): the ${app_id} macro is
            # substituted per replica by the scheduler at materialize time.
            for role in app.roles:
                role.env.setdefault(
                    "TORCHX_JOB_ID",
                    make_app_handle(scheduler, self._name, "${app_id}"),
                )

            # workspace build (mutates role.image for build-based scheds)
            if workspace:
                from torchx_amd.workspace.api import WorkspaceMixin

                if isinstance(sched, WorkspaceMixin):
                    sched.build_workspaces(app, workspace, cfg)

            return sched.submit_dryrun(app, cfg)

    def schedule(self, dryrun_info: AppDryRunInfo) -> AppHandle:
        scheduler = dryrun_info._scheduler
        assert scheduler, "dryrun_info must come from Runner.dryrun"
        sched = self._scheduler(scheduler)
        with log_event(
            "schedule", scheduler, self._name,
            app_image=(dryrun_info._app.roles[0].image
                       if dryrun_info._app and dryrun_info._app.roles else None),
        ) as ctx:
            app_id = sched.schedule(dryrun_info)
            ctx.event.app_id = app_id
        return make_app_handle(scheduler, self._name, app_id)

    # -- monitoring ---------------------------------------------------------
    def status(self, app_handle: AppHandle) -> Optional[AppStatus]:
        scheduler, _, app_id = parse_app_handle(app_handle)
        sched = self._scheduler(scheduler)
        with log_event("status", scheduler, self._name, app_id=app_id):
            desc = sched.describe(app_id)
        if desc is None:
            return None
        return AppStatus(
            state=desc.state,
            num_restarts=max(desc.num_restarts, 0),
            msg=desc.msg,
            structured_error_msg=desc.structured_error_msg,
            ui_url=desc.ui_url,
            roles=desc.roles_statuses,
        )

    def describe(self, app_handle: AppHandle) -> Optional[AppDef]:
        scheduler, _, app_id = parse_app_handle(app_handle)
        sched = self._scheduler(scheduler)
        with log_event("describe", scheduler, self._name, app_id=app_id):
            desc = sched.describe(app_id)
        if desc is None:
            return None
        return AppDef(name=app_id, roles=desc.roles)

    def wait(self, app_handle: AppHandle,
             wait_interval: float = 10.0) -> Optional[AppStatus]:
        while True:
            status = self.status(app_handle)
            if status is None or status.is_terminal():
                return status
            time.sleep(wait_interval)

    def cancel(self, app_handle: AppHandle) -> None:
        scheduler, _, app_id = parse_app_handle(app_handle)
        with log_event("cancel", scheduler, self._name, app_id=app_id):
            self._scheduler(scheduler).cancel(app_id)

    def delete(self, app_handle: AppHandle) -> None:
        scheduler, _, app_id = parse_app_handle(app_handle)
        with log_event("delete", scheduler, self._name, app_id=app_id):
            self._scheduler(scheduler).delete(app_id)

    def stop(self, app_handle: AppHandle) -> None:
        self.cancel(app_handle)

    def list(self, scheduler: str) -> List[ListAppResponse]:
        with log_event("list", scheduler, self._name):
            responses = self._scheduler(scheduler).list()
        for r in responses:
            r.app_handle = make_app_handle(scheduler, self._name, r.app_id)
        return responses

    def log_lines(
        self,
        app_handle: AppHandle,
        role_name: str,
        k: int = 0,
        regex: Optional[str] = None,
        since: Optional[datetime] = None,
        until: Optional[datetime] = None,
        should_tail: bool = False,
        streams: Optional[Stream] = None,
    ) -> Iterable[str]:
        scheduler, _, app_id = parse_app_handle(app_handle)
        with log_event("log_lines", scheduler, self._name, app_id=app_id):
            return self._scheduler(scheduler).log_iter(
                app_id, role_name, k, regex, since, until, should_tail,
                streams,
            )


def get_runner(
    name: Optional[str] = None,
    component_defaults: Optional[Dict[str, Dict[str, str]]] = None,
) -> Runner:
    return Runner(
        name=name or "torchx",
        component_defaults=component_defaults,
    )
