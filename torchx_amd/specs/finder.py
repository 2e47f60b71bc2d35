"""Component finder: resolves component identifiers to functions.

Resolution order (parity: torchx/specs/finder.py + runner/api.py:173-179):
  1. ``path/to/file.py:fn_name`` — exec the file, take fn_name
  2. ``module.path.fn_name`` dotted builtins under torchx_amd.components
     (e.g. ``dist.ddp`` or ``utils.echo``)
  3. entry-point group ``torchx_amd.components`` namespaces
"""

from __future__ import annotations

import importlib
import inspect
import os
import runpy
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional

from .api import AppDef


class ComponentNotFoundException(Exception):
    pass


class ComponentValidationException(Exception):
    pass


@dataclass
class _Component:
    name: str
    description: str
    fn_name: str
    fn: Callable[..., AppDef]


def _validate_component_fn(fn: Callable[..., Any], name: str) -> None:
    sig = inspect.signature(fn)
    ret = sig.return_annotation
    if ret is not inspect.Signature.empty:
        rname = getattr(ret, "__name__", None) or str(ret)
        if not rname.endswith("AppDef"):
            raise ComponentValidationException(
                f"component {name} must return AppDef, declares {ret}"
            )
    for pname, p in sig.parameters.items():
        if p.kind == inspect.Parameter.VAR_KEYWORD:
            raise ComponentValidationException(
                f"component {name}: **kwargs params are not supported"
            )


def _from_file(path: str, fn_name: str) -> _Component:
    if not os.path.isfile(path):
        raise ComponentNotFoundException(f"component file not found: {path}")
    # AST lint before exec: typed args + AppDef return (reference parity:
    # torchx/specs/finder.py:266 CustomComponentsFinder + file_linter)
    from .file_linter import validate as _lint

    msgs = _lint(path, fn_name)
    if any("not found" in m.description for m in msgs):
        raise ComponentNotFoundException(
            f"function {fn_name!r} not found in {path}"
        )
    errors = [m for m in msgs if m.severity == "error"]
    if errors:
        raise ComponentValidationException(
            "; ".join(f"{path}:{m.line}: {m.description}" for m in errors)
        )
    ns = runpy.run_path(path)
    fn = ns.get(fn_name)
    if fn is None or not callable(fn):
        raise ComponentNotFoundException(
            f"function {fn_name!r} not found in {path}"
        )
    _validate_component_fn(fn, f"{path}:{fn_name}")
    return _Component(
        name=f"{path}:{fn_name}",
        description=(inspect.getdoc(fn) or "").split("\n")[0],
        fn_name=fn_name,
        fn=fn,
    )


def _entry_point_modules() -> Dict[str, str]:
    out: Dict[str, str] = {}
    try:
        from importlib.metadata import entry_points

        eps = entry_points()
        # legacy reference group kept working for drop-in compatibility
        for gname in ("torchx_amd.components", "torchx.components"):
            group = (
                eps.select(group=gname)
                if hasattr(eps, "select")
                else eps.get(gname, [])
            )
            for ep in group:
                out.setdefault(ep.name, ep.value)
    except Exception:  # noqa: BLE001
        pass
    return out


def _from_module(name: str) -> _Component:
    mod_path, _, fn_name = name.rpartition(".")
    if not mod_path:
        raise ComponentNotFoundException(
            f"invalid component name {name!r}; expected module.fn or file.py:fn"
        )
    candidates = [f"torchx_amd.components.{mod_path}", mod_path]
    # entry-point namespaces: "ns.sub.fn" where ns is registered
    eps = _entry_point_modules()
    ns = name.split(".")[0]
    if ns in eps:
        rest = mod_path[len(ns):].lstrip(".")
        candidates.insert(0, eps[ns] + (f".{rest}" if rest else ""))
    last_err: Optional[Exception] = None
    for cand in candidates:
        try:
            mod = importlib.import_module(cand)
        except ImportError as e:
            last_err = e
            continue
        fn = getattr(mod, fn_name, None)
        if fn is not None and callable(fn):
            _validate_component_fn(fn, name)
            return _Component(
                name=name,
                description=(inspect.getdoc(fn) or "").split("\n")[0],
                fn_name=fn_name,
                fn=fn,
            )
    raise ComponentNotFoundException(
        f"component {name!r} not found (tried {candidates}): {last_err}"
    )


def get_component(name: str) -> _Component:
    if ":" in name:
        path, _, fn_name = name.rpartition(":")
        return _from_file(path, fn_name)
    return _from_module(name)


def get_components() -> Dict[str, _Component]:
    """All builtin components (walk torchx_amd.components modules)."""
    import torchx_amd.components as comps

    out: Dict[str, _Component] = {}
    pkg_dir = os.path.dirname(comps.__file__)
    for fname in sorted(os.listdir(pkg_dir)):
        if not fname.endswith(".py") or fname.startswith("_"):
            continue
        mod_name = fname[:-3]
        mod = importlib.import_module(f"torchx_amd.components.{mod_name}")
        for fn_name, fn in inspect.getmembers(mod, inspect.isfunction):
            if fn_name.startswith("_") or fn.__module__ != mod.__name__:
                continue
            sig = inspect.signature(fn)
            ret = sig.return_annotation
            rname = getattr(ret, "__name__", None) or str(ret)
            if not rname.endswith("AppDef"):
                continue
            full = f"{mod_name}.{fn_name}"
            out[full] = _Component(
                name=full,
                description=(inspect.getdoc(fn) or "").split("\n")[0],
                fn_name=fn_name,
                fn=fn,
            )
    return out


def get_builtin_source(name: str) -> str:
    comp = get_component(name)
    return inspect.getsource(comp.fn)
