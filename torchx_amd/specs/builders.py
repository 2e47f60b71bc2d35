"""Component-function -> AppDef materialization.

A *component* is a typed Python function returning an AppDef.  We build an
argparse parser from its signature (types + docstring help), parse the user's
string args, and call it (parity: torchx/specs/builders.py:30-270).
"""

from __future__ import annotations

import argparse
import inspect
import re
import typing
from typing import Any, Callable, Dict, List, Optional

from .api import AppDef, BindMount, DeviceMount, Mount, MountType, VolumeMount


class ComponentError(Exception):
    pass


def _decode_string(value: str, annotation: Any) -> Any:
    """Decode a CLI string into the annotated type."""
    origin = typing.get_origin(annotation)
    args = typing.get_args(annotation)
    if annotation in (inspect.Parameter.empty, str, Any, None):
        return value
    if annotation is bool:
        return value.strip().lower() in ("1", "true", "yes", "on")
    if annotation in (int, float):
        return annotation(value)
    if origin is typing.Union:  # Optional[T]
        non_none = [a for a in args if a is not type(None)]
        if value.lower() == "none":
            return None
        return _decode_string(value, non_none[0]) if non_none else value
    if origin in (list, List):
        elem = args[0] if args else str
        return [_decode_string(v, elem) for v in value.split(",") if v != ""]
    if origin in (dict, Dict):
        kt = args[0] if args else str
        vt = args[1] if len(args) > 1 else str
        out = {}
        for pair in value.split(","):
            if not pair:
                continue
            k, _, v = pair.partition("=")
            if not _:
                k, _, v = pair.partition(":")
            out[_decode_string(k, kt)] = _decode_string(v, vt)
        return out
    return value


def _docstring_param_help(fn: Callable[..., Any]) -> Dict[str, str]:
    """Extract ``Args:``-section parameter help from a Google/Sphinx docstring."""
    doc = inspect.getdoc(fn) or ""
    helps: Dict[str, str] = {}
    # Google style: "    name: help text"
    in_args = False
    current: Optional[str] = None
    for line in doc.splitlines():
        stripped = line.strip()
        if stripped.lower() in ("args:", "arguments:", "parameters:"):
            in_args = True
            continue
        if in_args:
            if not stripped or (stripped.endswith(":") and " " not in stripped):
                in_args = False
                current = None
                continue
            m = re.match(r"^([*\w]+)\s*(?:\([^)]*\))?\s*:\s*(.*)$", stripped)
            if m:
                current = m.group(1).lstrip("*")
                helps[current] = m.group(2)
            elif current:
                helps[current] += " " + stripped
    # Sphinx style  :param name: help
    for m in re.finditer(r":param\s+([*\w]+)\s*:\s*(.+)", doc):
        helps.setdefault(m.group(1).lstrip("*"), m.group(2).strip())
    return helps


def _fn_summary(fn: Callable[..., Any]) -> str:
    doc = inspect.getdoc(fn) or ""
    lines = []
    for line in doc.splitlines():
        if not line.strip():
            if lines:
                break
            continue
        lines.append(line.strip())
    return " ".join(lines)


def create_args_parser(fn: Callable[..., Any]) -> argparse.ArgumentParser:
    sig = inspect.signature(fn)
    helps = _docstring_param_help(fn)
    from .file_linter import ComponentHelpFormatter

    parser = argparse.ArgumentParser(
        prog=fn.__name__,
        description=_fn_summary(fn),
        formatter_class=ComponentHelpFormatter,
    )
    for name, p in sig.parameters.items():
        help_txt = helps.get(name, "")
        # single-char params also get a short flag (-j 1x2, -m mod, -h named
        # resource is taken by --help so it stays long-only); reference
        # parity: torchx/specs/builders.py:113 Annotated short flags
        flags = [f"--{name}"]
        if len(name) == 1 and name != "h":
            flags.append(f"-{name}")
        if p.kind == inspect.Parameter.VAR_POSITIONAL:
            parser.add_argument(name, nargs="*", default=[], help=help_txt)
        elif p.default is inspect.Parameter.empty:
            parser.add_argument(*flags, required=True, help=help_txt)
        else:
            parser.add_argument(*flags, default=p.default, help=help_txt)
    return parser


def materialize_appdef(
    fn: Callable[..., AppDef],
    args: List[str],
    defaults: Optional[Dict[str, str]] = None,
) -> AppDef:
    """Parse ``args`` against ``fn``'s signature and call it.

    Precedence: CLI args > ``defaults`` (e.g. from .torchxconfig) > function
    defaults (parity: torchx/specs/builders.py:137-179).
    """
    sig = inspect.signature(fn)
    try:
        hints = typing.get_type_hints(fn)
    except Exception:  # noqa: BLE001 — unresolvable forward refs
        hints = {}
    parser = create_args_parser(fn)
    if defaults:
        known = {
            f"--{k}": v
            for k, v in defaults.items()
            if k in sig.parameters
        }
        # inject config defaults for args not given on the CLI
        given = {a.split("=")[0] for a in args if a.startswith("--")}
        for flag, v in known.items():
            if flag not in given:
                args = [flag, v] + args
    has_varargs = any(
        p.kind == inspect.Parameter.VAR_POSITIONAL for p in sig.parameters.values()
    )
    extras: List[str] = []
    if has_varargs:
        # everything from the first token that is not a known `--param value`
        # pair belongs to *args, preserving order (REMAINDER semantics that
        # also allow `--known` flags before the tail)
        known_flags = set()
        for n, p in sig.parameters.items():
            if p.kind == inspect.Parameter.VAR_POSITIONAL:
                continue
            known_flags.add(f"--{n}")
            if len(n) == 1 and n != "h":
                known_flags.add(f"-{n}")
        head: List[str] = []
        i = 0
        while i < len(args):
            tok = args[i]
            name, eq, _ = tok.partition("=")
            if name in known_flags:
                if eq:
                    head.append(tok)
                    i += 1
                else:
                    head.extend(args[i : i + 2])
                    i += 2
            else:
                extras = args[i:]
                break
        ns = parser.parse_args(head)
    else:
        ns = parser.parse_args(args)

    # params before *args must be passed positionally; keyword-only after it
    pos_vals: List[Any] = []
    kw_vals: Dict[str, Any] = {}
    seen_varargs = False
    for name, p in sig.parameters.items():
        if p.kind == inspect.Parameter.VAR_POSITIONAL:
            pos_vals.extend((getattr(ns, name) or []) + extras)
            seen_varargs = True
            continue
        raw = getattr(ns, name)
        ann = hints.get(name, p.annotation)
        if isinstance(raw, str) and ann is not str:
            raw = _decode_string(raw, ann)
        if seen_varargs or p.kind == inspect.Parameter.KEYWORD_ONLY or not has_varargs:
            kw_vals[name] = raw
        else:
            pos_vals.append(raw)

    appdef = fn(*pos_vals, **kw_vals)
    if not isinstance(appdef, AppDef):
        raise ComponentError(
            f"component {fn.__name__} returned {type(appdef)}, expected AppDef"
        )
    return appdef


_MOUNT_TYPES = {t.value: t for t in MountType}


def parse_mounts(opts: List[str]) -> List[Mount]:
    """Parse ``type=bind,src=/x,dst=/y[,readonly]`` mount strings
    (parity: torchx/specs/builders.py:336)."""
    mounts: List[Mount] = []
    # each mount is one comma-joined group starting with type=
    group: Dict[str, str] = {}
    groups: List[Dict[str, str]] = []
    for opt in opts:
        for part in opt.split(","):
            if not part:
                continue
            k, _, v = part.partition("=")
            k = k.strip().lower()
            if k == "type" and group:
                groups.append(group)
                group = {}
            group[k] = v.strip() if _ else "true"
    if group:
        groups.append(group)

    for g in groups:
        mtype = _MOUNT_TYPES.get(g.get("type", ""))
        if mtype is None:
            raise ValueError(f"mount needs type=bind|volume|device, got {g}")
        ro = g.get("readonly", "false").lower() in ("true", "1", "")
        if mtype == MountType.BIND:
            mounts.append(BindMount(src_path=g["src"], dst_path=g["dst"], read_only=ro))
        elif mtype == MountType.VOLUME:
            mounts.append(VolumeMount(src=g["src"], dst_path=g["dst"], read_only=ro))
        else:
            mounts.append(
                DeviceMount(
                    src_path=g["src"],
                    dst_path=g.get("dst", g["src"]),
                    permissions=g.get("perm", "rwm"),
                )
            )
    return mounts
