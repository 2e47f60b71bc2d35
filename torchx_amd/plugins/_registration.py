"""@register decorators (parity: torchx/plugins/_registration.py:198-288),
including fractional named-resource generation (powers-of-two GPU splits and
halved-memory aliases)."""

from __future__ import annotations

from typing import Callable, Optional

from torchx_amd.specs.api import Resource

from ._registry import PluginType, registry


class register:
    @staticmethod
    def scheduler(name: str) -> Callable:
        def deco(factory: Callable) -> Callable:
            registry().add(PluginType.SCHEDULER, name, factory)
            return factory

        return deco

    @staticmethod
    def tracker(name: str) -> Callable:
        def deco(factory: Callable) -> Callable:
            registry().add(PluginType.TRACKER, name, factory)
            return factory

        return deco

    @staticmethod
    def named_resource(
        name: str,
        powers_of_two_gpus: bool = False,
        halve_mem_down_to: Optional[int] = None,
    ) -> Callable:
        def deco(factory: Callable[[], Resource]) -> Callable[[], Resource]:
            reg = registry()
            reg.add(PluginType.NAMED_RESOURCE, name, factory)
            # register into the spec library too
            from torchx_amd import specs

            specs.register_named_resource(name, factory)
            base = factory()
            if powers_of_two_gpus and base.gpu > 1:
                g = base.gpu // 2
                while g >= 1:
                    frac = g / base.gpu
                    res = Resource(
                        cpu=max(1, int(base.cpu * frac)),
                        gpu=g,
                        memMB=int(base.memMB * frac),
                        capabilities=dict(base.capabilities),
                    )
                    alias = f"{name}_{g}"
                    specs.register_named_resource(alias, lambda r=res: r)
                    reg.add(PluginType.NAMED_RESOURCE, alias, lambda r=res: r)
                    g //= 2
            if halve_mem_down_to:
                mem = base.memMB // 2
                factor = 0.5
                while mem >= halve_mem_down_to:
                    res = Resource(
                        cpu=base.cpu, gpu=base.gpu, memMB=mem,
                        capabilities=dict(base.capabilities),
                    )
                    alias = f"{name}_{factor}x"
                    specs.register_named_resource(alias, lambda r=res: r)
                    reg.add(PluginType.NAMED_RESOURCE, alias, lambda r=res: r)
                    mem //= 2
                    factor /= 2
            return factory

        return deco
