"""Docker workspace (mocked client), workspace walking/ignore patterns,
macro substitution, structured error files, and runopts parsing edges
(parity: torchx workspace/api_test, specs/api_test)."""

import json
import os
from unittest.mock import MagicMock

import pytest

from torchx_amd.specs import AppDef, Resource, Role, macros, runopts
from torchx_amd.specs.api import read_structured_error


class TestWalkWorkspace:
    def _mk(self, tmp_path, files):
        for rel, content in files.items():
            p = tmp_path / rel
            p.parent.mkdir(parents=True, exist_ok=True)
            p.write_text(content)

    def test_ignore_patterns(self, tmp_path):
        from torchx_amd.workspace.api import walk_workspace

        self._mk(tmp_path, {
            "main.py": "x",
            "data/big.bin": "x",
            "keep/me.txt": "x",
            ".torchxignore": "data/\n*.log\n!keep.log\n",
            "run.log": "x",
            "keep.log": "x",
        })
        rels = {rel for _, rel in walk_workspace(str(tmp_path))}
        assert "main.py" in rels
        assert "keep/me.txt" in rels
        assert "keep.log" in rels        # negated pattern
        assert "run.log" not in rels
        assert "data/big.bin" not in rels


class TestDockerWorkspace:
    def _mixin(self, client):
        from torchx_amd.workspace.docker_workspace import DockerWorkspaceMixin

        return DockerWorkspaceMixin(docker_client=client)

    def test_build_updates_role_image(self, tmp_path):
        (tmp_path / "main.py").write_text("print('hi')")
        client = MagicMock()
        client.api.build.return_value = [
            {"stream": "Step 1/2"},
            {"aux": {"ID": "sha256:abc123"}},
        ]
        ws = self._mixin(client)
        role = Role(name="r", image="base:latest", entrypoint="python")
        ws.build_workspace_and_update_role(role, str(tmp_path), {})
        assert role.image == "sha256:abc123"
        kwargs = client.api.build.call_args.kwargs
        assert kwargs["buildargs"]["IMAGE"] == "base:latest"
        assert kwargs["custom_context"]

    def test_build_error_raises(self, tmp_path):
        from torchx_amd.workspace.docker_workspace import BuildError

        client = MagicMock()
        client.api.build.return_value = [{"error": "boom"}]
        ws = self._mixin(client)
        role = Role(name="r", image="img", entrypoint="x")
        with pytest.raises(BuildError):
            ws.build_workspace_and_update_role(role, str(tmp_path), {})

    def test_push_images_flow(self):
        client = MagicMock()
        ws = self._mixin(client)
        app = AppDef(name="a", roles=[
            Role(name="r", image="sha256:deadbeef", entrypoint="x"),
        ])
        to_push = ws.dryrun_push_images(app, {"image_repo": "example.com/repo"})
        assert app.roles[0].image == "example.com/repo:deadbeef"
        assert to_push == {"sha256:deadbeef": ("example.com/repo", "deadbeef")}
        ws.push_images(to_push)
        client.images.get.assert_called_with("sha256:deadbeef")

    def test_push_without_repo_raises(self):
        ws = self._mixin(MagicMock())
        app = AppDef(name="a", roles=[
            Role(name="r", image="sha256:ff", entrypoint="x"),
        ])
        with pytest.raises(KeyError):
            ws.dryrun_push_images(app, {})

    def test_default_dockerfile_in_context(self, tmp_path):
        import tarfile

        from torchx_amd.workspace.docker_workspace import _build_context

        (tmp_path / "app.py").write_text("pass")
        ctx = _build_context("img", str(tmp_path))
        with tarfile.open(fileobj=ctx) as tf:
            names = tf.getnames()
        assert "Dockerfile.torchx" in names and "app.py" in names


class TestMacros:
    def test_substitution(self):
        role = Role(
            name="r", image=macros.img_root, entrypoint="run",
            args=["--id", macros.app_id, "--replica", macros.replica_id],
            env={"RANK0": macros.rank0_env},
        )
        values = macros.Values(img_root="/img", app_id="app_1",
                               replica_id="3", rank0_env="HOST0")
        out = values.apply(role)
        # image is NOT substituted (handled by the scheduler's
        # ImageProvider, as in the reference)
        assert out.args == ["--id", "app_1", "--replica", "3"]
        assert out.env["RANK0"] == "HOST0"
        # original untouched
        assert role.args[1] == macros.app_id


class TestStructuredError:
    def test_torchelastic_nested_schema(self, tmp_path):
        ef = tmp_path / "err.json"
        ef.write_text(json.dumps({
            "message": {
                "message": "RuntimeError: boom",
                "extraInfo": {"py_callstack": "Traceback ..."},
            }
        }))
        msg = read_structured_error(str(ef))
        assert msg and "boom" in msg

    def test_flat_schema(self, tmp_path):
        ef = tmp_path / "err.json"
        ef.write_text(json.dumps({"message": "plain failure"}))
        assert "plain failure" in read_structured_error(str(ef))

    def test_missing_file(self):
        assert read_structured_error("/nope/err.json") is None


class TestRunopts:
    def _opts(self):
        o = runopts()
        o.add("log_dir", type_=str, help="h")
        o.add("count", type_=int, default=1, help="h")
        o.add("flags", type_=list, help="h")
        o.add("mapping", type_=dict, help="h")
        o.add("enabled", type_=bool, default=False, help="h")
        return o

    def test_cfg_from_str(self):
        cfg = self._opts().cfg_from_str(
            "log_dir=/tmp/x,count=3,flags=a;b;c,enabled=True"
        )
        assert cfg["log_dir"] == "/tmp/x"
        assert cfg["count"] == 3
        assert cfg["flags"] == ["a", "b", "c"]
        assert cfg["enabled"] is True

    def test_dict_parsing(self):
        cfg = self._opts().cfg_from_str("mapping=k1:v1,k2:v2")
        assert cfg["mapping"] == {"k1": "v1", "k2": "v2"}

    def test_resolve_defaults_and_unknown_passthrough(self):
        o = self._opts()
        r = o.resolve({"log_dir": "/l"})
        assert r["count"] == 1
        # unknown keys pass through (reference parity: resolve starts
        # from {**cfg}; schedulers may accept extras)
        assert o.resolve({"extra": 1})["extra"] == 1


class TestRunoptsBoolVocabulary:
    def test_ini_bool_vocabulary(self):
        # reference parity: INI bool vocabulary (specs/api.py:1079-1325)
        o = runopts()
        o.add("b", type_=bool, default=False, help="h")
        for v in ("True", "true", "1", "yes"):
            assert o.cfg_from_str(f"b={v}")["b"] is True, v
        for v in ("False", "false", "0", "no"):
            assert o.cfg_from_str(f"b={v}")["b"] is False, v
