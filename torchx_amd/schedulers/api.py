"""Scheduler abstraction: dryrun-first submit protocol.

Parity with the reference Scheduler ABC (torchx/schedulers/api.py:373):
``submit = submit_dryrun + schedule``; ``describe/cancel/delete/list/
log_iter`` monitor the data plane.  Run-config schemas are ``runopts``
(torchx_amd.specs.api).
"""

from __future__ import annotations

import abc
import inspect
import re
import typing
from dataclasses import MISSING, dataclass, field, fields, is_dataclass
from datetime import datetime
from enum import Enum
from typing import (
    Any, Dict, Generic, Iterable, Iterator, List, Mapping, Optional, TypeVar,
)

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


# ---------------------------------------------------------------------------
# StructuredOpts: typed scheduler config (parity: torchx/schedulers/api.py:79)
# ---------------------------------------------------------------------------


def _snake_to_camel(name: str) -> str:
    head, *rest = name.split("_")
    return head + "".join(p.title() for p in rest)


def _camel_to_snake(name: str) -> str:
    return re.sub(r"(?<=[a-z0-9])([A-Z])", r"_\1", name).lower()


def _unwrap_optional(tp: Any) -> Any:
    """Optional[X] / X | None -> X."""
    if typing.get_origin(tp) is typing.Union:
        args = [a for a in typing.get_args(tp) if a is not type(None)]
        if len(args) == 1:
            return args[0]
    return tp


def _is_opts_type(tp: Any) -> bool:
    return (isinstance(tp, type) and issubclass(tp, StructuredOpts)
            and is_dataclass(tp))


@dataclass
class StructuredOpts(Mapping[str, Any]):
    """Declare scheduler run options as a ``@dataclass``: field types and
    defaults become the ``runopts`` schema, field docstrings become the
    help text, nested ``StructuredOpts`` fields flatten to dot-prefixed
    keys (``k8s.context``), snake_case fields accept camelCase aliases,
    and a ``cfg_key`` metadata entry overrides the external key for names
    that cannot be identifiers (``mail-user``)::

        @dataclass
        class MyOpts(StructuredOpts):
            cluster: str
            '''cluster to submit to'''
            retries: int = 3
            '''number of retry attempts'''

    ``MyOpts.as_runopts()`` feeds ``torchx runopts``/``torchx configure``;
    ``MyOpts.from_cfg(cfg)`` gives typed access in ``_submit_dryrun``.
    The Mapping protocol keeps instances usable where a cfg dict is
    expected.
    """

    # -- construction -------------------------------------------------------
    @classmethod
    def from_cfg(cls, cfg: Mapping[str, Any]):
        hints = typing.get_type_hints(cls)
        kwargs: Dict[str, Any] = {}
        for f in fields(cls):
            ftype = _unwrap_optional(hints.get(f.name, str))
            if _is_opts_type(ftype):
                prefix = f.name + "."
                nested = {k[len(prefix):]: v for k, v in cfg.items()
                          if k.startswith(prefix)}
                if nested or (f.default is MISSING
                              and f.default_factory is MISSING):
                    kwargs[f.name] = ftype.from_cfg(nested)
                continue
            key = f.metadata.get("cfg_key", f.name)
            for cand in (key, f.name, _snake_to_camel(f.name)):
                if cand in cfg and cfg[cand] is not None:
                    kwargs[f.name] = cfg[cand]
                    break
        return cls(**kwargs)

    # -- runopts schema ------------------------------------------------------
    @classmethod
    def field_docstrings(cls) -> Dict[str, str]:
        """Attribute docstrings (the string literal following each field)."""
        import ast
        import textwrap

        docs: Dict[str, str] = {}
        for klass in reversed(cls.__mro__):
            if not is_dataclass(klass) or klass is StructuredOpts:
                continue
            try:
                src = textwrap.dedent(inspect.getsource(klass))
                body = ast.parse(src).body[0].body
            except (OSError, TypeError, SyntaxError, IndexError):
                continue
            for stmt, nxt in zip(body, body[1:]):
                if (isinstance(stmt, ast.AnnAssign)
                        and isinstance(stmt.target, ast.Name)
                        and isinstance(nxt, ast.Expr)
                        and isinstance(nxt.value, ast.Constant)
                        and isinstance(nxt.value.value, str)):
                    docs[stmt.target.id] = nxt.value.value.strip()
        hints = typing.get_type_hints(cls)
        for f in fields(cls):
            ftype = _unwrap_optional(hints.get(f.name, str))
            if _is_opts_type(ftype):
                for k, d in ftype.field_docstrings().items():
                    docs[f"{f.name}.{k}"] = d
        return docs

    @classmethod
    def as_runopts(cls) -> runopts:
        opts = runopts()
        hints = typing.get_type_hints(cls)
        docs = cls.field_docstrings()
        for f in fields(cls):
            ftype = _unwrap_optional(hints.get(f.name, str))
            if _is_opts_type(ftype):
                for key, sub in ftype.as_runopts():
                    opts.add(f"{f.name}.{key}", type_=sub.opt_type,
                             default=sub.default, required=sub.required,
                             help=sub.help)
                continue
            has_default = (f.default is not MISSING
                           or f.default_factory is not MISSING)
            default = f.default if f.default is not MISSING else None
            opts.add(
                f.metadata.get("cfg_key", f.name),
                type_=ftype,
                default=default,
                required=not has_default,
                help=docs.get(f.name, f.name),
            )
        return opts

    # -- Mapping protocol (dict-style compatibility) -------------------------
    def __getitem__(self, key: str) -> Any:
        if "." in key:
            head, rest = key.split(".", 1)
            nested = getattr(self, _camel_to_snake(head), None)
            if isinstance(nested, StructuredOpts):
                return nested[rest]
            raise KeyError(key)
        snake = _camel_to_snake(key)
        names = {f.name for f in fields(self)}
        if snake in names:
            return getattr(self, snake)
        for f in fields(self):
            if f.metadata.get("cfg_key") == key:
                return getattr(self, f.name)
        raise KeyError(key)

    def __iter__(self) -> Iterator[str]:
        hints = typing.get_type_hints(type(self))
        for f in fields(self):
            ftype = _unwrap_optional(hints.get(f.name, str))
            if _is_opts_type(ftype):
                nested = getattr(self, f.name)
                if isinstance(nested, StructuredOpts):
                    for k in nested:
                        yield f"{f.name}.{k}"
            else:
                yield f.metadata.get("cfg_key", f.name)

    def __len__(self) -> int:
        return sum(1 for _ in self)

    def __or__(self, other: "StructuredOpts") -> Dict[str, Any]:
        merged: Dict[str, Any] = {k: self[k] for k in self}
        merged.update({k: other[k] for k in other})
        return merged


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
