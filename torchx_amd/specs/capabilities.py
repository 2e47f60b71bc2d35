"""Typed, namespaced accessors into ``Resource.capabilities``
(reference parity: torchx/specs/capabilities.py:30).

MI355X builds ship keys for xGMI link count and HBM size so schedulers
can make topology-aware placement decisions.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Generic, Optional, Type, TypeVar

from .api import Resource

V = TypeVar("V")


@dataclass(frozen=True)
class CapabilityKey(Generic[V]):
    """A typed key ``namespace.name`` into ``Resource.capabilities``."""

    namespace: str
    name: str
    type: Type[V]

    @property
    def key(self) -> str:
        return f"{self.namespace}.{self.name}"

    def set(self, resource: Resource, value: V) -> None:
        resource.capabilities[self.key] = value

    def get(self, resource: Resource,
            default: Optional[V] = None) -> Optional[V]:
        """Value under :attr:`key`, or ``default``; TypeError on a type
        mismatch (bool does not satisfy an int-typed key)."""
        if self.key not in resource.capabilities:
            return default
        value = resource.capabilities[self.key]
        bool_for_int = self.type is int and type(value) is bool
        if bool_for_int or not isinstance(value, self.type):
            raise TypeError(
                f"capability `{self.key}` expected `{self.type.__name__}`, "
                f"got `{type(value).__name__}`: {value!r}"
            )
        return value


# MI355X topology capabilities (consumed by schedulers for affinity)
XGMI_LINKS = CapabilityKey("amd", "xgmi_links_per_gpu", int)
HBM_GB = CapabilityKey("amd", "hbm_gb_per_gpu", int)
GFX_ARCH = CapabilityKey("amd", "gfx_arch", str)
