"""Tests for the small parity modules that had no dedicated suite:
capabilities, metadata_keys, deprecations, pipelines namespace, version.
(reference parity: torchx/specs/test/capabilities_test.py,
torchx/test/deprecations_test.py, torchx/test/version_test.py)
"""

import importlib
import warnings

import pytest

from torchx_amd import deprecations, pipelines, version
from torchx_amd.specs import capabilities, metadata_keys
from torchx_amd.specs.api import Resource

# `torchx_amd.specs.named_resources` the *attribute* is the accessor
# function; the module itself must be imported explicitly.
named_resources = importlib.import_module("torchx_amd.specs.named_resources")


def _res() -> Resource:
    return Resource(cpu=1, gpu=0, memMB=1024)


class TestCapabilities:
    def test_set_get_roundtrip(self):
        r = _res()
        capabilities.XGMI_LINKS.set(r, 7)
        assert r.capabilities["amd.xgmi_links_per_gpu"] == 7
        assert capabilities.XGMI_LINKS.get(r) == 7

    def test_get_default_when_absent(self):
        r = _res()
        assert capabilities.HBM_GB.get(r) is None
        assert capabilities.HBM_GB.get(r, 288) == 288

    def test_type_mismatch_raises(self):
        r = _res()
        r.capabilities[capabilities.GFX_ARCH.key] = 950  # int into str key
        with pytest.raises(TypeError):
            capabilities.GFX_ARCH.get(r)

    def test_bool_does_not_satisfy_int_key(self):
        r = _res()
        r.capabilities[capabilities.XGMI_LINKS.key] = True
        with pytest.raises(TypeError):
            capabilities.XGMI_LINKS.get(r)

    def test_mi355x_named_resources_stamped(self):
        # every mi355x.* named resource carries the topology capabilities
        for name, factory in named_resources.NAMED_RESOURCES.items():
            if not name.startswith("mi355x."):
                continue
            r = factory()
            assert capabilities.GFX_ARCH.get(r) == "gfx950", name
            assert capabilities.HBM_GB.get(r) == 288, name
            assert capabilities.XGMI_LINKS.get(r) == 7, name


class TestMetadataKeys:
    def test_keys_are_wire_stable(self):
        # these strings cross the launcher<->scheduler boundary: frozen
        assert metadata_keys.CONTEXT == "torchx/context"
        assert metadata_keys.VERSION == "torchx/version"
        assert metadata_keys.EXPERIMENT_NAME == "torchx/experiment-name"
        assert metadata_keys.RUN_NAME == "torchx/run-name"

    def test_app_metadata(self):
        md = metadata_keys.app_metadata("cli_run", "0.1.0")
        assert md == {"torchx/context": "cli_run", "torchx/version": "0.1.0"}


class TestDeprecations:
    def test_deprecated_warns_once(self):
        @deprecations.deprecated(replacement="new_fn")
        def old_fn(x):
            return x + 1

        with warnings.catch_warnings(record=True) as w:
            warnings.simplefilter("always")
            assert old_fn(1) == 2
            assert old_fn(2) == 3
        dep = [x for x in w if issubclass(x.category, DeprecationWarning)]
        assert len(dep) == 1
        assert "new_fn" in str(dep[0].message)

    def test_deprecated_preserves_identity(self):
        @deprecations.deprecated()
        def some_fn():
            """doc"""

        assert some_fn.__name__ == "some_fn"
        assert some_fn.__doc__ == "doc"

    def test_deprecated_module_warns(self):
        with warnings.catch_warnings(record=True) as w:
            warnings.simplefilter("always")
            deprecations.deprecated_module("torchx.old", "torchx_amd.new")
        assert any("torchx.old" in str(x.message) for x in w)


class TestVersionAndPipelines:
    def test_version_string(self):
        assert version.__version__
        assert version.TORCHX_IMAGE.endswith(version.__version__)

    def test_image_env_override(self, monkeypatch):
        monkeypatch.setenv("TORCHX_IMAGE", "registry.local/custom:tag")
        assert version.get_torchx_image() == "registry.local/custom:tag"

    def test_image_default(self, monkeypatch):
        monkeypatch.delenv("TORCHX_IMAGE", raising=False)
        assert version.get_torchx_image() == version.TORCHX_IMAGE

    def test_pipelines_namespace_importable(self):
        # intentionally empty in-core; adapters ship as plugins
        mod = importlib.import_module("torchx_amd.pipelines")
        assert mod is pipelines


class TestIds:
    def test_make_unique_sanitizes_and_suffixes(self):
        from torchx_amd.schedulers.ids import make_unique, random_id

        uid = make_unique("my app/v2")
        base, _, suffix = uid.rpartition("-")
        assert base == "my-app-v2"
        assert len(suffix) == 8 and suffix.isalnum()
        # ids are unique across calls
        assert make_unique("x") != make_unique("x")
        # degenerate names still produce a valid id
        assert make_unique("///").startswith("app-")
        assert len(random_id(4)) == 4


def test_settings_constants_wire_stable():
    from torchx_amd import settings as s

    assert s.ENV_TORCHX_JOB_ID == "TORCHX_JOB_ID"
    assert s.ENV_TORCHX_TRACKERS == "TORCHX_TRACKERS"
    assert s.ENV_TORCHXCONFIG == "TORCHXCONFIG"
    assert s.ENV_TORCHX_RANK0_HOST == "TORCHX_RANK0_HOST"
    assert s.ENV_HIP_VISIBLE_DEVICES == "HIP_VISIBLE_DEVICES"


def test_notebook_workspace_roundtrip():
    from torchx_amd.notebook import get_workspace, write_workspace_file

    path = write_workspace_file("pkg/mod.py", "VALUE = 41\n")
    assert path.startswith(get_workspace())
    import fsspec

    fs, p = fsspec.core.url_to_fs(path)
    with fs.open(p) as f:
        assert b"VALUE = 41" in f.read()
