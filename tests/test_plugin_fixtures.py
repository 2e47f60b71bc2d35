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
