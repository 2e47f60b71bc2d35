"""Plugin system (parity: torchx/plugins/): two discovery channels —
``torchx_amd_plugins.*`` namespace packages and entry-point groups
(``torchx_amd.schedulers`` / ``torchx_amd.tracker`` /
``torchx_amd.named_resources``) — plus ``@register`` decorators."""

from ._registry import PluginRegistry, PluginType, registry  # noqa: F401
from ._registration import register  # noqa: F401
