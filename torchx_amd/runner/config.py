""".torchxconfig INI support (parity: torchx/runner/config.py).

Sections:
  [scheduler]        — per-scheduler runopt defaults (e.g. [local_cwd])
  [component:NAME]   — component param defaults (e.g. [component:dist.ddp])
  [cli:CMD]          — CLI arg defaults (e.g. [cli:run])
  [torchx:tracker] / [tracker:NAME] — tracker wiring

Discovery: $TORCHXCONFIG overrides; else $HOME/.torchxconfig then
$CWD/.torchxconfig (later files win per key: precedence CLI > env-file >
home > cwd > code defaults — reference config.py:84-91)."""

from __future__ import annotations

import configparser
import os
from pathlib import Path
from typing import Any, Dict, List, Mapping, Optional

CONFIG_FILE = ".torchxconfig"
ENV_TORCHXCONFIG = "TORCHXCONFIG"

def find_configs(dirs: Optional[List[str]] = None) -> List[str]:
    env = os.environ.get(ENV_TORCHXCONFIG)
    if env:
        return [env] if os.path.isfile(env) else []
    if dirs is None:
        dirs = [str(Path.home()), os.getcwd()]
    out = []
    for d in dirs:
        p = os.path.join(d, CONFIG_FILE)
        if os.path.isfile(p):
            out.append(p)
    return out


def _read(dirs: Optional[List[str]] = None) -> configparser.ConfigParser:
    cp = configparser.ConfigParser()
    cp.optionxform = str  # preserve case
    for path in find_configs(dirs):
        cp.read(path)
    return cp


def load_sections(prefix: str,
                  dirs: Optional[List[str]] = None) -> Dict[str, Dict[str, str]]:
    """All sections named ``prefix:*`` -> {suffix: {k: v}}."""
    cp = _read(dirs)
    out: Dict[str, Dict[str, str]] = {}
    for section in cp.sections():
        if section.startswith(prefix + ":"):
            out[section[len(prefix) + 1:]] = dict(cp[section])
    return out


def get_config(prefix: str, name: str, key: str,
               dirs: Optional[List[str]] = None) -> Optional[str]:
    cp = _read(dirs)
    section = f"{prefix}:{name}" if name else prefix
    if cp.has_section(section) and key in cp[section]:
        return cp[section][key]
    return None


def load(scheduler: str, cfg: Dict[str, Any],
         dirs: Optional[List[str]] = None) -> None:
    """Overlay [scheduler-name] section values onto ``cfg`` (only keys not
    already present — CLI wins)."""
    cp = _read(dirs)
    if not cp.has_section(scheduler):
        return
    for k, v in cp[scheduler].items():
        if k not in cfg or cfg[k] is None:
            cfg[k] = v


def apply(scheduler: str, cfg: Dict[str, Any],
          dirs: Optional[List[str]] = None) -> None:
    load(scheduler, cfg, dirs)


def dump(path: str, schedulers: Optional[List[str]] = None,
         required_only: bool = False) -> None:
    """Write a .torchxconfig template generated from each scheduler's
    actual runopts (``torchx configure``; reference config.py:222 —
    optional opts pre-filled with defaults, required ones as #FIXME).
    Orchestrator plugins registered under the
    ``torchx_amd.schedulers.orchestrator`` entry-point group are
    included alongside the built-in schedulers."""
    from torchx_amd.schedulers import get_scheduler_factories

    factories = dict(get_scheduler_factories())
    try:
        from importlib.metadata import entry_points

        eps = entry_points()
        found = (eps.select(group="torchx_amd.schedulers.orchestrator")
                 if hasattr(eps, "select")
                 else eps.get("torchx_amd.schedulers.orchestrator", []))
        for ep in found:
            factories.setdefault(ep.name, ep.load())
    except Exception:  # noqa: BLE001
        pass
    explicit = schedulers is not None and len(schedulers) > 0
    names = schedulers if explicit else list(factories)
    lines = ["#", "# torchx_amd configuration (torchx configure)",
             "# Fill in the #FIXME values; delete what you don't need.",
             "#"]
    for name in names:
        factory = factories.get(name)
        if factory is None:
            raise ValueError(
                f"unknown scheduler {name!r}; known: {sorted(factories)}"
            )
        try:
            opts = factory("_").run_opts()
        except Exception:  # noqa: BLE001 — a broken (plugin) scheduler
            # must not break `torchx configure` for the others
            if explicit:
                raise
            continue
        lines.append(f"[{name}]")
        for key, opt in opts:
            if opt.required:
                lines.append(f"{key} = #FIXME ({opt.opt_type.__name__}) "
                             f"{opt.help}")
            elif not required_only:
                default = "" if opt.default is None else opt.default
                lines.append(f"# {key} = {default}")
        lines.append("")
    with open(path, "w") as f:
        f.write("\n".join(lines) + "\n")


def get_configured_trackers(
    dirs: Optional[List[str]] = None,
) -> Dict[str, Optional[str]]:
    """[torchx:tracker] name = config  (+ [tracker:NAME] config = ...)."""
    cp = _read(dirs)
    out: Dict[str, Optional[str]] = {}
    if cp.has_section("torchx:tracker"):
        for name, cfgval in cp["torchx:tracker"].items():
            out[name] = cfgval or None
    for name in list(out):
        sec = f"tracker:{name}"
        if cp.has_section(sec) and "config" in cp[sec]:
            out[name] = cp[sec]["config"]
    return out
