"""Version info (parity: torchx/version.py).

``TORCHX_IMAGE_TAG``/entry-point override lets deployments pin the default
container image used by components when none is given.
"""

import os

__version__ = "0.1.0"

# default container image for docker/k8s components when unset
TORCHX_IMAGE = f"torchx-amd:{__version__}"


def get_torchx_image() -> str:
    env = os.environ.get("TORCHX_IMAGE")
    if env:
        return env
    try:
        from importlib.metadata import entry_points

        eps = entry_points()
        found = (eps.select(group="torchx_amd.version")
                 if hasattr(eps, "select")
                 else eps.get("torchx_amd.version", []))
        for ep in found:
            if ep.name == "get_torchx_image":
                return ep.load()()
    except Exception:  # noqa: BLE001
        pass
    return TORCHX_IMAGE
