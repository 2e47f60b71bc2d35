"""Plugin registry (parity: torchx/plugins/_registry.py:37-753).

Discovery:
  1. ``torchx_amd_plugins.{schedulers,named_resources,tracker}`` namespace
     packages (implicit namespace dirs on sys.path included);
  2. entry points ``torchx_amd.schedulers`` / ``torchx_amd.tracker`` /
     ``torchx_amd.named_resources``;
  3. imperative ``@register.*`` decorators (_registration.py).

Duplicate registration of the SAME factory is idempotent; a different
factory under an existing name raises.
"""

from __future__ import annotations

import importlib
import logging
import pkgutil
import threading
from enum import Enum
from typing import Any, Callable, Dict, List, Optional

log = logging.getLogger(__name__)


class PluginType(str, Enum):
    SCHEDULER = "scheduler"
    NAMED_RESOURCE = "named_resource"
    TRACKER = "tracker"


class DuplicatePluginError(Exception):
    pass


class PluginRegistry:
    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._plugins: Dict[PluginType, Dict[str, Callable]] = {
            t: {} for t in PluginType
        }
        self._scanned = False
        self._errors: List[str] = []

    # -- registration -------------------------------------------------------
    def add(self, ptype: PluginType, name: str, factory: Callable) -> None:
        with self._lock:
            existing = self._plugins[ptype].get(name)
            if existing is not None and existing is not factory:
                raise DuplicatePluginError(
                    f"{ptype.value} plugin {name!r} already registered"
                )
            self._plugins[ptype][name] = factory

    # -- discovery ----------------------------------------------------------
    def _scan(self) -> None:
        if self._scanned:
            return
        self._scanned = True
        # namespace packages
        try:
            import torchx_amd_plugins  # type: ignore[import-not-found]

            for sub in ("schedulers", "named_resources", "tracker"):
                try:
                    pkg = importlib.import_module(f"torchx_amd_plugins.{sub}")
                except ImportError:
                    continue
                for m in pkgutil.iter_modules(pkg.__path__):
                    try:
                        importlib.import_module(
                            f"torchx_amd_plugins.{sub}.{m.name}"
                        )
                    except Exception as e:  # noqa: BLE001
                        self._errors.append(
                            f"torchx_amd_plugins.{sub}.{m.name}: {e}"
                        )
        except ImportError:
            pass
        except Exception as e:  # noqa: BLE001 — a broken plugin ROOT
            # (reference's broken_root fixture) must not crash discovery
            self._errors.append(f"torchx_amd_plugins: {e}")
        # entry points
        try:
            from importlib.metadata import entry_points

            eps = entry_points()
            for group, ptype in (
                ("torchx_amd.schedulers", PluginType.SCHEDULER),
                ("torchx_amd.tracker", PluginType.TRACKER),
                ("torchx_amd.named_resources", PluginType.NAMED_RESOURCE),
                # legacy reference group names keep existing torchx
                # plugin packages working unchanged (wire-contract parity)
                ("torchx.schedulers", PluginType.SCHEDULER),
                ("torchx.tracker", PluginType.TRACKER),
                ("torchx.named_resources", PluginType.NAMED_RESOURCE),
            ):
                found = (
                    eps.select(group=group)
                    if hasattr(eps, "select")
                    else eps.get(group, [])
                )
                for ep in found:
                    try:
                        self.add(ptype, ep.name, ep.load())
                    except Exception as e:  # noqa: BLE001
                        self._errors.append(f"{group}:{ep.name}: {e}")
        except Exception as e:  # noqa: BLE001
            self._errors.append(str(e))

    # -- queries ------------------------------------------------------------
    def scheduler_factories(self) -> Dict[str, Callable]:
        self._scan()
        return dict(self._plugins[PluginType.SCHEDULER])

    def tracker_factory(self, name: str) -> Optional[Callable]:
        self._scan()
        return self._plugins[PluginType.TRACKER].get(name)

    def named_resources(self) -> Dict[str, Callable]:
        self._scan()
        return dict(self._plugins[PluginType.NAMED_RESOURCE])

    def diagnostics(self) -> str:
        self._scan()
        lines = ["plugin registry:"]
        for t in PluginType:
            for name in sorted(self._plugins[t]):
                lines.append(f"  {t.value}: {name}")
        for e in self._errors:
            lines.append(f"  error: {e}")
        return "\n".join(lines)


_REGISTRY: Optional[PluginRegistry] = None
_REG_LOCK = threading.Lock()


def registry() -> PluginRegistry:
    global _REGISTRY
    with _REG_LOCK:
        if _REGISTRY is None:
            _REGISTRY = PluginRegistry()
        return _REGISTRY
