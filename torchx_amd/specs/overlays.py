"""Request overlays: patch a scheduler's native submit request with fields
not representable in :class:`AppDef`/:class:`Role`.

Behavior parity with the reference (torchx/specs/overlays.py:150-768):
``set_overlay`` stores overlays under ``target.metadata[namespace][kind]``
(accumulating: dicts merge, lists append); ``apply_overlay`` deep-merges an
overlay onto a scheduler request dict in place; ``PUT``/``JOIN``/``DEL``
operator keys override per-field behavior (replace / strategic-merge-by-key
/ remove); overlays may also be loaded from a JSON or YAML file URI.
"""

from __future__ import annotations

import copy
import json
import logging
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Set, Union

from .api import AppDef, Role

logger = logging.getLogger(__name__)

Overlay = Dict[str, Any]

_FORMAT_KEY = "__torchx_overlays__"
_PUT = "__put__:"
_JOIN = "__join__:"
_DEL = "__del__:"


def PUT(key: str) -> str:
    """Operator key: replace the base value entirely (no merge/append)."""
    return f"{_PUT}{key}"


def JOIN(key: str, *, on: str) -> str:
    """Operator key: strategic-merge list items matching on the ``on`` field;
    unmatched overlay items are appended."""
    return f"{_JOIN}{key}:{on}"


def DEL(key: str) -> str:
    """Operator key: remove the field from the base dict (value ignored)."""
    return f"{_DEL}{key}"


def _field_of(key: str) -> str:
    """Logical field name of a possibly operator-encoded key."""
    for prefix in (_PUT, _DEL):
        if key.startswith(prefix):
            return key[len(prefix):]
    if key.startswith(_JOIN):
        return key[len(_JOIN):].split(":", 1)[0]
    return key


def _strategic_merge(base_list: List[Any], overlay_list: List[Any],
                     merge_key: str) -> None:
    by_key = {
        item[merge_key]: item
        for item in base_list
        if isinstance(item, dict) and merge_key in item
    }
    for item in overlay_list:
        if isinstance(item, dict) and item.get(merge_key) in by_key:
            target = by_key[item[merge_key]]
            for k, v in item.items():
                if k != merge_key:
                    target[k] = copy.deepcopy(v)
        else:
            base_list.append(copy.deepcopy(item))


def _drop_field(base: Overlay, field: str, keep_plain: bool = False) -> None:
    for k in [k for k in base if _field_of(k) == field]:
        if keep_plain and k == field:
            continue
        del base[k]


def _apply_join(base: Overlay, key: str, value: Any, field: str) -> None:
    _drop_field(base, field, keep_plain=True)
    parts = key[len(_JOIN):].split(":", 1)
    if len(parts) != 2 or not parts[1]:
        raise ValueError(f"malformed JOIN key `{key}`; use JOIN(field, on=key)")
    merge_key = parts[1]
    if not (isinstance(value, list) and all(isinstance(i, dict) for i in value)):
        raise TypeError(f"JOIN overlay for `{field}` must be a list of dicts")
    if field in base:
        bv = base[field]
        if not (isinstance(bv, list) and all(isinstance(i, dict) for i in bv)):
            raise TypeError(f"JOIN base `{field}` must be a list of dicts")
        _strategic_merge(bv, value, merge_key)
    else:
        base[field] = copy.deepcopy(value)


def _merge_value(base: Overlay, key: str, value: Any, resolve: bool) -> None:
    if key in base:
        bv = base[key]
        if isinstance(bv, dict) and isinstance(value, dict):
            apply_overlay(bv, value, _resolve=resolve)
        elif isinstance(bv, list) and isinstance(value, list):
            bv.extend(copy.deepcopy(value))
        elif isinstance(bv, (dict, list)) or isinstance(value, (dict, list)):
            raise TypeError(
                f"type mismatch for `{key}`: "
                f"{type(bv).__name__} != {type(value).__name__}"
            )
        else:
            base[key] = value
    else:
        base[key] = copy.deepcopy(value)


def apply_overlay(base: Overlay, overlay: Overlay, *,
                  _resolve: bool = True) -> None:
    """Merge ``overlay`` into ``base`` in place.

    Defaults: dicts merge recursively, lists append, primitives overwrite.
    ``PUT``/``JOIN``/``DEL`` keys override per field. With ``_resolve=False``
    (accumulation mode used by :func:`set_overlay`) operator keys are stored
    verbatim; for a given field the last stored operation wins.
    """
    for key, value in overlay.items():
        field = _field_of(key)
        if _resolve:
            if key.startswith(_DEL):
                _drop_field(base, field)
                continue
            if key.startswith(_PUT):
                _drop_field(base, field)
                base[field] = copy.deepcopy(value)
                continue
            if key.startswith(_JOIN):
                _apply_join(base, key, value, field)
                continue
        # a new op (or plain set) for a field supersedes earlier ops
        for k in [k for k in base if k != key and _field_of(k) == field]:
            del base[k]
        _merge_value(base, key, value, _resolve)


def load_overlay_file(uri: str) -> Overlay:
    """Load an overlay dict from a local path / fsspec URI (JSON, else YAML)."""
    if "://" not in uri:
        uri = f"file://{uri}"
    import fsspec

    with fsspec.open(uri, "r") as f:
        contents = f.read()
    try:
        data = json.loads(contents)
    except json.JSONDecodeError:
        import yaml

        data = yaml.safe_load(contents)
    if not isinstance(data, dict):
        raise ValueError(f"overlay file `{uri}` must contain a dict")
    return data


def set_overlay(target: Union[AppDef, Role], namespace: str, kind: str,
                overlay: Overlay) -> None:
    """Store an overlay under ``target.metadata[namespace][kind]``.

    Repeated calls for the same (namespace, kind) accumulate with
    :func:`apply_overlay` semantics (operators stored unresolved).
    """
    if kind == _FORMAT_KEY:
        raise ValueError(f"overlay kind `{kind}` is reserved")
    md: Dict[str, Any] = target.metadata
    ns = md.setdefault(namespace, {})
    if not isinstance(ns, dict):
        ns = {}
        md[namespace] = ns
    ns[_FORMAT_KEY] = True
    existing = ns.setdefault(kind, {})
    if not isinstance(existing, dict):
        existing = {}
        ns[kind] = existing
    apply_overlay(existing, overlay, _resolve=False)


def get_overlay(target: Union[AppDef, Role], namespace: str,
                kind: str) -> Overlay:
    """Retrieve the overlay stored for (namespace, kind); ``{}`` if absent.
    A string metadata value is loaded as a file URI (JSON/YAML)."""
    if kind == _FORMAT_KEY:
        raise ValueError(f"overlay kind `{kind}` is reserved")
    ns = target.metadata.get(namespace)
    if ns is None:
        return {}
    if isinstance(ns, str):
        ns = load_overlay_file(ns)
    if not isinstance(ns, dict):
        return {}
    overlay = ns.get(kind)
    return overlay if isinstance(overlay, dict) else {}


def validate_overlay(overlay: Overlay, *,
                     blocklist: Optional[Sequence[str]] = None,
                     forbidden_keys: Optional[Set[str]] = None,
                     overlay_name: str = "overlay",
                     suggestion: str = "") -> None:
    """Reject overlays containing keys the scheduler owns (set them on the
    Role/AppDef instead) or keys belonging to a different overlay kind."""
    if blocklist:
        bad = [_field_of(k) for k in overlay if _field_of(k) in blocklist]
        if bad:
            keys = ", ".join(f"`{overlay_name}.{k}`" for k in bad)
            raise ValueError(
                f"disallowed overlay attributes {keys}: set them directly on "
                f"the role's attributes"
            )
    if forbidden_keys:
        misplaced = {_field_of(k) for k in overlay} & forbidden_keys
        if misplaced:
            msg = f"{overlay_name} overlay contains misplaced keys: {misplaced}."
            if suggestion:
                msg = f"{msg} {suggestion}"
            raise ValueError(msg)


@dataclass(frozen=True)
class OverlaySpec:
    """A scheduler's overlay surface declared once: namespace, kind and the
    keys users must set via Role/AppDef attributes instead."""

    namespace: str
    kind: str
    blocklist: Sequence[str] = ()

    def set(self, target: Union[AppDef, Role], overlay: Overlay) -> None:
        validate_overlay(overlay, blocklist=list(self.blocklist),
                         overlay_name=self.kind)
        set_overlay(target, self.namespace, self.kind, overlay)

    def get(self, target: Union[AppDef, Role]) -> Overlay:
        overlay = get_overlay(target, self.namespace, self.kind)
        validate_overlay(overlay, blocklist=list(self.blocklist),
                         overlay_name=self.kind)
        return overlay
