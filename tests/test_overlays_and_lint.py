"""Tests for request overlays (PUT/JOIN/DEL), the component AST linter,
and structured args (parity: torchx/specs/overlays.py, file_linter.py,
components/structured_arg.py)."""

import json
import textwrap

import pytest

from torchx_amd.specs import AppDef, Role
from torchx_amd.specs.overlays import (
    DEL,
    JOIN,
    PUT,
    OverlaySpec,
    apply_overlay,
    get_overlay,
    load_overlay_file,
    set_overlay,
    validate_overlay,
)


def _role():
    return Role(name="trainer", image="img", entrypoint="train.py")


class TestApplyOverlay:
    def test_dict_merge_list_append_primitive_overwrite(self):
        base = {"spec": {"cpu": "500m"}, "tags": ["prod"], "replicas": 1}
        apply_overlay(base, {"spec": {"memory": "1Gi"}, "tags": ["gpu"],
                             "replicas": 3})
        assert base == {"spec": {"cpu": "500m", "memory": "1Gi"},
                        "tags": ["prod", "gpu"], "replicas": 3}

    def test_put_replaces(self):
        base = {"containers": [{"name": "old1"}, {"name": "old2"}]}
        apply_overlay(base, {PUT("containers"): [{"name": "only"}]})
        assert base == {"containers": [{"name": "only"}]}

    def test_join_strategic_merge(self):
        base = {"containers": [{"name": "main", "image": "v1", "cpu": "1"}]}
        apply_overlay(base, {JOIN("containers", on="name"): [
            {"name": "main", "memory": "1Gi"},
            {"name": "sidecar", "image": "proxy"},
        ]})
        assert base["containers"] == [
            {"name": "main", "image": "v1", "cpu": "1", "memory": "1Gi"},
            {"name": "sidecar", "image": "proxy"},
        ]

    def test_del_removes(self):
        base = {"keep": 1, "remove_me": "old"}
        apply_overlay(base, {DEL("remove_me"): None})
        assert base == {"keep": 1}

    def test_none_is_not_del(self):
        base = {"a": 1}
        apply_overlay(base, {"a": None})
        assert base == {"a": None}

    def test_type_mismatch_raises(self):
        with pytest.raises(TypeError):
            apply_overlay({"a": [1]}, {"a": "x"})

    def test_join_requires_list_of_dicts(self):
        with pytest.raises(TypeError):
            apply_overlay({"c": [{"name": "x"}]},
                          {JOIN("c", on="name"): ["notadict"]})


class TestSetGetOverlay:
    def test_accumulate(self):
        role = _role()
        set_overlay(role, "kubernetes", "V1Pod",
                    {"spec": {"nodeSelector": {"accel": "mi355x"}}})
        set_overlay(role, "kubernetes", "V1Pod",
                    {"spec": {"tolerations": [{"key": "gpu"}]}})
        got = get_overlay(role, "kubernetes", "V1Pod")
        assert got == {"spec": {"nodeSelector": {"accel": "mi355x"},
                                "tolerations": [{"key": "gpu"}]}}

    def test_kind_isolation(self):
        role = _role()
        set_overlay(role, "ns", "A", {"x": 1})
        assert get_overlay(role, "ns", "B") == {}

    def test_missing_returns_empty(self):
        assert get_overlay(_role(), "nope", "K") == {}

    def test_operators_last_wins_in_accumulation(self):
        role = _role()
        set_overlay(role, "ns", "K", {PUT("f"): [1]})
        set_overlay(role, "ns", "K", {"f": [2]})
        base = {"f": [0]}
        apply_overlay(base, get_overlay(role, "ns", "K"))
        assert base == {"f": [0, 2]}  # plain set superseded the PUT

    def test_file_uri_overlay(self, tmp_path):
        p = tmp_path / "ov.json"
        p.write_text(json.dumps({"K": {"x": 1}}))
        role = _role()
        role.metadata["ns"] = str(p)
        assert get_overlay(role, "ns", "K") == {"x": 1}

    def test_load_yaml(self, tmp_path):
        p = tmp_path / "ov.yaml"
        p.write_text("spec:\n  a: 1\n")
        assert load_overlay_file(str(p)) == {"spec": {"a": 1}}

    def test_appdef_target(self):
        app = AppDef(name="a", roles=[_role()])
        set_overlay(app, "ns", "K", {"x": 1})
        assert get_overlay(app, "ns", "K") == {"x": 1}


class TestValidateOverlay:
    def test_blocklist(self):
        with pytest.raises(ValueError, match="env"):
            validate_overlay({"env": {"A": "b"}}, blocklist=["env"],
                             overlay_name="PodSpec")

    def test_blocklist_sees_through_operators(self):
        with pytest.raises(ValueError):
            validate_overlay({PUT("env"): {}}, blocklist=["env"])

    def test_overlay_spec_roundtrip(self):
        spec = OverlaySpec("kubernetes", "V1Pod", blocklist=("command",))
        role = _role()
        spec.set(role, {"spec": {"nodeSelector": {"gpu": "true"}}})
        assert spec.get(role) == {"spec": {"nodeSelector": {"gpu": "true"}}}
        with pytest.raises(ValueError):
            spec.set(role, {"command": ["x"]})


class TestK8sPodOverlay:
    def test_overlay_applied_to_pod(self):
        from torchx_amd.schedulers.kubernetes_scheduler import (
            KubernetesScheduler,
        )
        from torchx_amd.specs import Resource

        role = Role(name="train", image="img", entrypoint="bash",
                    resource=Resource(cpu=2, gpu=0, memMB=1024),
                    num_replicas=1)
        set_overlay(role, "kubernetes", "V1Pod",
                    {"spec": {"nodeSelector": {"pool": "mi355x"}}})
        app = AppDef(name="app", roles=[role])
        sched = KubernetesScheduler("test")
        info = sched.submit_dryrun(app, {"queue": "default"})
        tasks = info.request.resource["spec"]["tasks"]
        pod = tasks[0]["template"]
        assert pod["spec"]["nodeSelector"] == {"pool": "mi355x"}


class TestLinter:
    def _lint(self, tmp_path, src, fn="comp"):
        from torchx_amd.specs.file_linter import validate

        f = tmp_path / "comp.py"
        f.write_text(textwrap.dedent(src))
        return validate(str(f), fn)

    def test_valid_component(self, tmp_path):
        msgs = self._lint(tmp_path, '''
            from typing import Dict, List, Optional
            from torchx_amd.specs import AppDef
            def comp(x: int, name: str = "a", env: Optional[Dict[str, str]] = None,
                     *args: str) -> AppDef:
                """My comp.

                Args:
                    x: the x
                """
                return AppDef(name=name, roles=[])
        ''')
        assert [m for m in msgs if m.severity == "error"] == []

    def test_missing_annotation(self, tmp_path):
        msgs = self._lint(tmp_path, '''
            from torchx_amd.specs import AppDef
            def comp(x) -> AppDef:
                """d"""
                return AppDef(name="a", roles=[])
        ''')
        assert any("missing type annotation" in m.description for m in msgs)

    def test_bad_type(self, tmp_path):
        msgs = self._lint(tmp_path, '''
            from torchx_amd.specs import AppDef
            def comp(x: object) -> AppDef:
                """d"""
                return AppDef(name="a", roles=[])
        ''')
        assert any("unsupported type" in m.description for m in msgs)

    def test_missing_return(self, tmp_path):
        msgs = self._lint(tmp_path, '''
            def comp(x: int):
                """d"""
                return None
        ''')
        assert any("must declare" in m.description for m in msgs)

    def test_missing_docstring_is_warning(self, tmp_path):
        msgs = self._lint(tmp_path, '''
            from torchx_amd.specs import AppDef
            def comp(x: int) -> AppDef:
                return AppDef(name="a", roles=[])
        ''')
        assert [m.severity for m in msgs
                if "docstring" in m.description] == ["warning"]

    def test_finder_rejects_bad_component(self, tmp_path):
        from torchx_amd.specs.finder import (
            ComponentValidationException,
            get_component,
        )

        f = tmp_path / "bad.py"
        f.write_text("def comp(x):\n    return None\n")
        with pytest.raises(ComponentValidationException):
            get_component(f"{f}:comp")


class TestStructuredArg:
    def test_name_both(self):
        from torchx_amd.components.structured_arg import StructuredNameArgument

        a = StructuredNameArgument.parse_from("foo/bar", script="x/y.py")
        assert (a.experiment_name, a.run_name) == ("foo", "bar")

    def test_name_exp_only_derives_run(self):
        from torchx_amd.components.structured_arg import StructuredNameArgument

        a = StructuredNameArgument.parse_from("foo/", script="x/baz.py")
        assert (a.experiment_name, a.run_name) == ("foo", "baz")
        a = StructuredNameArgument.parse_from("foo/", m="a.b.mod")
        assert a.run_name == "mod"

    def test_name_run_only(self):
        from torchx_amd.components.structured_arg import StructuredNameArgument

        a = StructuredNameArgument.parse_from("/bar", m="a.b")
        assert (a.experiment_name, a.run_name) == ("default-experiment", "bar")
        a = StructuredNameArgument.parse_from("bar", m="a.b")
        assert a.run_name == "bar"

    def test_name_requires_one_of_m_script(self):
        from torchx_amd.components.structured_arg import StructuredNameArgument

        with pytest.raises(ValueError):
            StructuredNameArgument.parse_from("foo/bar")
        with pytest.raises(ValueError):
            StructuredNameArgument.parse_from("foo/bar", m="a", script="b.py")

    def test_j_infers_nproc_from_gpus(self):
        from torchx_amd.components.structured_arg import StructuredJArgument

        a = StructuredJArgument.parse_from(h="mi355x.8gpu", j="2")
        assert (a.nnodes, a.nproc_per_node) == (2, 8)
        assert str(a) == "2x8"

    def test_j_explicit(self):
        from torchx_amd.components.structured_arg import StructuredJArgument

        with pytest.warns(UserWarning):
            a = StructuredJArgument.parse_from(h="mi355x.8gpu", j="2x4")
        assert (a.nnodes, a.nproc_per_node) == (2, 4)

    def test_j_cpu_host_requires_nproc(self):
        from torchx_amd.components.structured_arg import StructuredJArgument

        with pytest.raises(ValueError):
            StructuredJArgument.parse_from(h="cpu.medium", j="2")
