"""Plugin namespace-package discovery with good and broken fixture plugins
(parity: the reference's plugins/test/{default,bad,broken_root} error-path
fixtures — broken plugins must be collected as diagnostics, never crash
discovery)."""

import subprocess
import sys
import textwrap
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

DRIVER = """
import sys
sys.path.insert(0, {repo!r})
sys.path.insert(0, {fixtures!r})
from torchx_amd.plugins import registry

reg = registry()
scheds = reg.scheduler_factories()
assert "fixture_sched" in scheds, scheds
diag = reg.diagnostics()
assert "boom-on-import" in diag or "broken" in diag, diag
print("PLUGIN_FIXTURES_OK")
"""


def test_namespace_plugins_good_and_broken(tmp_path):
    # implicit namespace package: torchx_amd_plugins/schedulers/{good,broken}
    pkg = tmp_path / "torchx_amd_plugins" / "schedulers"
    pkg.mkdir(parents=True)
    (pkg / "good.py").write_text(textwrap.dedent("""
        from torchx_amd.plugins import register

        @register.scheduler("fixture_sched")
        def make(session_name, **kw):
            return None
    """))
    (pkg / "broken.py").write_text(
        'raise RuntimeError("boom-on-import")\n'
    )
    out = subprocess.run(
        [sys.executable, "-c",
         DRIVER.format(repo=str(REPO), fixtures=str(tmp_path))],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    assert "PLUGIN_FIXTURES_OK" in out.stdout


BROKEN_ROOT_DRIVER = """
import sys
sys.path.insert(0, {repo!r})
sys.path.insert(0, {fixtures!r})
from torchx_amd.plugins import registry

reg = registry()
# discovery must survive a plugin ROOT whose import raises
scheds = reg.scheduler_factories()
assert scheds == {{}}, scheds
assert "root-boom" in reg.diagnostics(), reg.diagnostics()
print("BROKEN_ROOT_OK")
"""


def test_broken_plugin_root_does_not_crash_discovery(tmp_path):
    # reference parity: plugins/test broken_root fixture — the namespace
    # root itself fails at import with a non-ImportError
    pkg = tmp_path / "torchx_amd_plugins"
    pkg.mkdir()
    (pkg / "__init__.py").write_text('raise RuntimeError("root-boom")\n')
    out = subprocess.run(
        [sys.executable, "-c",
         BROKEN_ROOT_DRIVER.format(repo=str(REPO), fixtures=str(tmp_path))],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    assert "BROKEN_ROOT_OK" in out.stdout


MULTI_TYPE_DRIVER = """
import sys
sys.path.insert(0, {repo!r})
sys.path.insert(0, {fixtures!r})
from torchx_amd.plugins import registry

reg = registry()
assert reg.tracker_factory("fixture_tracker") is not None
res = reg.named_resources()
assert "fixture.res" in res, res
r = res["fixture.res"]()
assert r.gpu == 2, r
print("MULTI_TYPE_OK")
"""


def test_tracker_and_named_resource_namespace_plugins(tmp_path):
    root = tmp_path / "torchx_amd_plugins"
    (root / "tracker").mkdir(parents=True)
    (root / "tracker" / "fix.py").write_text(textwrap.dedent("""
        from torchx_amd.plugins import register

        @register.tracker("fixture_tracker")
        def make(config=None):
            return None
    """))
    (root / "named_resources").mkdir()
    (root / "named_resources" / "fix.py").write_text(textwrap.dedent("""
        from torchx_amd.plugins import register
        from torchx_amd.specs.api import Resource

        @register.named_resource("fixture.res")
        def make():
            return Resource(cpu=8, gpu=2, memMB=1024)
    """))
    out = subprocess.run(
        [sys.executable, "-c",
         MULTI_TYPE_DRIVER.format(repo=str(REPO), fixtures=str(tmp_path))],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    assert "MULTI_TYPE_OK" in out.stdout


def test_duplicate_registration_rules():
    # in-process: same factory is idempotent, different factory raises
    from torchx_amd.plugins._registry import (
        DuplicatePluginError,
        PluginRegistry,
        PluginType,
    )

    reg = PluginRegistry()

    def factory_a():
        return None

    def factory_b():
        return None

    reg.add(PluginType.SCHEDULER, "dup", factory_a)
    reg.add(PluginType.SCHEDULER, "dup", factory_a)  # idempotent
    try:
        reg.add(PluginType.SCHEDULER, "dup", factory_b)
        raise AssertionError("expected DuplicatePluginError")
    except DuplicatePluginError:
        pass
    # same name in a DIFFERENT plugin type is fine
    reg.add(PluginType.TRACKER, "dup", factory_b)
