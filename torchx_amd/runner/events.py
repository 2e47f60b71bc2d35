"""Structured telemetry for runner API calls (parity:
torchx/runner/events/: TorchxEvent api.py:24, log_event __init__.py:81).

Every public Runner call is wrapped in :func:`log_event`, recording api
name, scheduler, app id/image, runcfg, cpu/wall time and exception
details, then emitted through a pluggable logging handler ("null" default,
"console" prints to stderr)."""

from __future__ import annotations

import json
import logging
import sys
import time
import traceback
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, Optional

_HANDLERS: Dict[str, logging.Handler] = {}


def register_handler(name: str, handler: logging.Handler) -> None:
    _HANDLERS[name] = handler


def get_logging_handler(dest: str = "null") -> logging.Handler:
    if dest in _HANDLERS:
        return _HANDLERS[dest]
    if dest == "console":
        return logging.StreamHandler(sys.stderr)
    return logging.NullHandler()


_logger: Optional[logging.Logger] = None


def _trace_logger(dest: str = "null") -> logging.Logger:
    global _logger
    if _logger is None:
        _logger = logging.getLogger("torchx_amd.events")
        _logger.setLevel(logging.INFO)
        _logger.propagate = False
        _logger.addHandler(get_logging_handler(dest))
    return _logger


@dataclass
class TorchxEvent:
    session: str
    scheduler: str
    api: str
    app_id: Optional[str] = None
    app_image: Optional[str] = None
    runcfg: Optional[str] = None
    workspace: Optional[str] = None
    source: str = "UNKNOWN"
    cpu_time_usec: Optional[int] = None
    wall_time_usec: Optional[int] = None
    start_epoch_time_usec: Optional[int] = None
    raw_exception: Optional[str] = None
    exception_type: Optional[str] = None
    exception_message: Optional[str] = None

    def serialize(self) -> str:
        return json.dumps(asdict(self))


def record(event: TorchxEvent, dest: str = "null") -> None:
    _trace_logger(dest).info(event.serialize())


class log_event:
    """Context manager recording one runner API call."""

    def __init__(self, api: str, scheduler: str = "", session: str = "",
                 app_id: Optional[str] = None,
                 app_image: Optional[str] = None,
                 runcfg: Optional[str] = None,
                 workspace: Optional[str] = None) -> None:
        self._event = TorchxEvent(
            session=session, scheduler=scheduler, api=api, app_id=app_id,
            app_image=app_image, runcfg=runcfg, workspace=workspace,
            start_epoch_time_usec=int(time.time() * 1e6),
        )
        self._t0 = 0.0
        self._c0 = 0.0

    def __enter__(self) -> "log_event":
        self._t0 = time.perf_counter()
        self._c0 = time.process_time()
        return self

    @property
    def event(self) -> TorchxEvent:
        return self._event

    def __exit__(self, exc_type, exc, tb) -> None:
        self._event.wall_time_usec = int((time.perf_counter() - self._t0) * 1e6)
        self._event.cpu_time_usec = int((time.process_time() - self._c0) * 1e6)
        if exc is not None:
            self._event.exception_type = exc_type.__name__
            self._event.exception_message = str(exc)
            self._event.raw_exception = "".join(
                traceback.format_exception(exc_type, exc, tb)
            )
            if tb is not None:
                frame = traceback.extract_tb(tb)[-1]
                self._event.source = f"{frame.filename}:{frame.lineno}"
        record(self._event)
