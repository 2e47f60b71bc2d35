"""Spec-layer unit tests (model: torchx/specs/test/api_test.py behaviors)."""

from typing import Dict, List, Optional

import pytest

from torchx_amd.specs import (
    AppDef,
    AppState,
    AppStatus,
    BindMount,
    DeviceMount,
    InvalidRunConfigException,
    Resource,
    RetryPolicy,
    Role,
    VolumeMount,
    is_terminal,
    macros,
    make_app_handle,
    materialize_appdef,
    named_resources,
    parse_app_handle,
    parse_mounts,
    resource,
    runopts,
)


def test_resource_named_mi355x():
    r = resource(h="mi355x.8gpu")
    assert r.gpu == 8
    assert r.capabilities["amd.com/hbm_gb"] == 288
    half = resource(h="mi355x.4gpu")
    assert half.gpu == 4
    assert half.memMB == r.memMB // 2
    assert half.cpu == r.cpu // 2


def test_resource_defaults():
    r = resource()
    assert (r.cpu, r.gpu, r.memMB) == (2, 0, 1024)


def test_resource_unknown_named():
    with pytest.raises(ValueError):
        resource(h="nope.xl")


def test_named_resources_has_generic():
    lib = named_resources()
    assert "gpu.xlarge" in lib and lib["gpu.xlarge"]().gpu == 8


def test_macros_apply_per_replica():
    role = Role(
        name="trainer",
        image="/img",
        entrypoint="bash",
        args=["-c", f"echo {macros.replica_id} {macros.app_id}"],
        env={"RANK0": macros.rank0_env},
    )
    values = macros.Values(
        img_root="/img", app_id="app_1", replica_id="3", rank0_env="TORCHX_RANK0_HOST"
    )
    rep = values.apply(role)
    assert rep.args == ["-c", "echo 3 app_1"]
    assert rep.env["RANK0"] == "TORCHX_RANK0_HOST"
    # original untouched
    assert macros.replica_id in role.args[1]


def test_app_handle_roundtrip():
    h = make_app_handle("local_cwd", "sess", "app_123")
    assert parse_app_handle(h) == ("local_cwd", "sess", "app_123")
    with pytest.raises(ValueError):
        parse_app_handle("not a handle")


def test_app_state_terminal():
    assert is_terminal(AppState.SUCCEEDED)
    assert is_terminal(AppState.FAILED)
    assert not is_terminal(AppState.RUNNING)
    assert AppStatus(state=AppState.CANCELLED).is_terminal()


def test_appstatus_error_format():
    s = AppStatus(
        state=AppState.FAILED,
        structured_error_msg='{"message": {"message": "boom", "extraInfo": {"py_callstack": "tb", "timestamp": "100"}}}',
    )
    out = s.format()
    assert "boom" in out and "tb" in out


def test_runopts_parse_resolve():
    opts = runopts()
    opts.add("log_dir", type_=str, help="dir", default="/tmp")
    opts.add("num", type_=int, help="n", required=True)
    opts.add("flag", type_=bool, help="b", default=False)
    opts.add("names", type_=List[str], help="l")
    opts.add("env", type_=Dict[str, str], help="d")

    cfg = opts.cfg_from_str("num=3,flag=True,names=a,b,c;env=K:V")
    resolved = opts.resolve(cfg)
    assert resolved["num"] == 3
    assert resolved["flag"] is True
    assert resolved["names"] == ["a", "b", "c"]
    assert resolved["env"] == {"K": "V"}
    assert resolved["log_dir"] == "/tmp"

    with pytest.raises(InvalidRunConfigException):
        opts.resolve({})


def test_materialize_appdef():
    def comp(name: str, replicas: int = 2, flags: Optional[List[str]] = None) -> AppDef:
        """My component.

        Args:
            name: app name
            replicas: replica count
            flags: extra flags
        """
        return AppDef(
            name=name,
            roles=[Role(name="r", image="/", num_replicas=replicas, args=flags or [])],
        )

    app = materialize_appdef(comp, ["--name", "x", "--replicas", "4", "--flags", "a,b"])
    assert app.name == "x"
    assert app.roles[0].num_replicas == 4
    assert app.roles[0].args == ["a", "b"]

    # config defaults apply when CLI omits them
    app2 = materialize_appdef(comp, ["--name", "y"], defaults={"replicas": "8"})
    assert app2.roles[0].num_replicas == 8


def test_materialize_varargs():
    def comp(script: str, *script_args: str) -> AppDef:
        return AppDef(name="s", roles=[Role(name="r", image="/", args=[script, *script_args])])

    app = materialize_appdef(comp, ["--script", "main.py", "-x", "1"])
    assert app.roles[0].args == ["main.py", "-x", "1"]


def test_parse_mounts():
    mounts = parse_mounts(
        ["type=bind,src=/a,dst=/b,readonly", "type=volume,src=vol,dst=/v", "type=device,src=/dev/kfd"]
    )
    assert isinstance(mounts[0], BindMount) and mounts[0].read_only
    assert isinstance(mounts[1], VolumeMount) and mounts[1].src == "vol"
    assert isinstance(mounts[2], DeviceMount) and mounts[2].dst_path == "/dev/kfd"


def test_retry_policy_str():
    assert str(RetryPolicy.REPLICA) == "REPLICA"
