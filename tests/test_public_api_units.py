"""Direct unit tests for public-API helpers that were previously covered
only through integration paths (parity anchors in each test docstring)."""

import torch


def test_parse_nnodes_forms():
    """reference components/dist.py:340 — "N", "NxM", "lo:hixM"."""
    from torchx_amd.components.dist import parse_nnodes

    assert parse_nnodes("2") == (2, 2, 1, "2:2")
    assert parse_nnodes("2x8") == (2, 2, 8, "2:2")
    assert parse_nnodes("1:4x8") == (1, 4, 8, "1:4")


def test_replace_role_and_appdef_to_dict():
    """reference specs/api.py — role copy with overrides; AppDef dict
    dump with stringified retry policy."""
    from torchx_amd.specs import AppDef, Resource, Role
    from torchx_amd.specs.api import appdef_to_dict, replace_role

    r = Role(name="w", image="/img", entrypoint="python",
             args=["-c", "1"], env={"A": "1"}, num_replicas=2,
             resource=Resource(cpu=2, gpu=1, memMB=1024))
    r2 = replace_role(r, num_replicas=4, name="x")
    assert (r2.name, r2.num_replicas) == ("x", 4)
    assert r.num_replicas == 2                       # original untouched
    assert r2.args is not r.args and r2.env is not r.env

    d = appdef_to_dict(AppDef(name="app", roles=[r]))
    assert d["name"] == "app"
    assert isinstance(d["roles"][0]["retry_policy"], str)


def test_mount_type_and_get_type_name():
    from torchx_amd.specs.api import MountType, get_type_name

    assert MountType("bind") is MountType.BIND
    assert get_type_name(int) == "int"
    assert "str" in get_type_name(str)


def test_filter_regex_and_split_lines():
    """reference schedulers/api.py:541-567 log helpers."""
    from torchx_amd.schedulers.api import (
        filter_regex, split_lines, split_lines_iterator,
    )

    lines = ["error: boom\n", "ok\n", "ERROR again\n"]
    assert list(filter_regex("error", lines)) == ["error: boom\n"]
    assert split_lines("a\nb\nc") == ["a\n", "b\n", "c"]
    # iterator form re-chunks arbitrary splits into whole lines
    out = list(split_lines_iterator(["a\nb", "c\n", "d"]))
    assert out == ["a\n", "bc\n", "d"]


def test_merge_workspace(tmp_path):
    """reference workspace/api.py:149-154 — later projects win."""
    from torchx_amd.specs.api import Workspace
    from torchx_amd.workspace.api import merge_workspace

    a, b = tmp_path / "a", tmp_path / "b"
    a.mkdir(), b.mkdir()
    (a / "x.txt").write_text("from-a")
    (a / "only_a.txt").write_text("a")
    (b / "x.txt").write_text("from-b")
    out = tmp_path / "out"
    out.mkdir()
    ws = Workspace(projects={str(a): "", str(b): ""})
    merge_workspace(ws, str(out))
    assert (out / "x.txt").read_text() == "from-b"   # later wins
    assert (out / "only_a.txt").read_text() == "a"


def test_find_configs_env_override(tmp_path, monkeypatch):
    """reference runner/config.py:484 — TORCHXCONFIG env takes over."""
    from torchx_amd.runner.config import find_configs

    cfg = tmp_path / ".torchxconfig"
    cfg.write_text("[local_cwd]\n")
    monkeypatch.setenv("TORCHXCONFIG", str(cfg))
    assert find_configs() == [str(cfg)]
    monkeypatch.setenv("TORCHXCONFIG", str(tmp_path / "missing"))
    assert find_configs() == []


def test_distributed_helpers(monkeypatch):
    """reference torchx/distributed — env-safe helpers without a pg."""
    from torchx_amd.distributed import (
        is_torchelastic_launched, local_device, on_local_rank0_first,
    )

    monkeypatch.delenv("RANK", raising=False)
    monkeypatch.delenv("TORCHELASTIC_RUN_ID", raising=False)
    assert not is_torchelastic_launched()
    monkeypatch.setenv("RANK", "0")
    monkeypatch.setenv("WORLD_SIZE", "1")
    assert is_torchelastic_launched()
    assert local_device() == torch.device("cpu")  # no GPU in CI
    with on_local_rank0_first():                  # world 1: no barrier
        pass


def test_trackers_from_environ_empty_and_unknown():
    """reference tracker/api.py:229-239 — env-driven tracker build."""
    from torchx_amd.tracker.api import trackers_from_environ

    assert trackers_from_environ({}) == []
    # unknown backends are skipped with a warning, not fatal (a worker
    # must not die because a tracker plugin is missing)
    assert trackers_from_environ({"TORCHX_TRACKERS": "no_such_backend"}) == []


def test_trackers_from_environ_fsspec(tmp_path):
    from torchx_amd.tracker.api import trackers_from_environ

    env = {
        "TORCHX_TRACKERS": "fsspec",
        "TORCHX_TRACKER_FSSPEC_CONFIG": str(tmp_path),
    }
    ts = trackers_from_environ(env)
    assert len(ts) == 1


def test_get_device_mounts_count_aware():
    """reference schedulers/devices.py:43-54 — per-count EFA/Neuron
    mounts, unknown names warn and skip, AMD GPU nodes are static."""
    import warnings as _w

    from torchx_amd.schedulers.devices import get_device_mounts

    m = get_device_mounts({"vpc.amazonaws.com/efa": 2})
    assert [d.src_path for d in m] == ["/dev/infiniband/uverbs0",
                                       "/dev/infiniband/uverbs1"]
    m = get_device_mounts({"aws.amazon.com/neurondevice": 1})
    assert m[0].src_path == "/dev/neuron0"
    m = get_device_mounts({"amd.com/gpu": 8})
    assert {d.src_path for d in m} == {"/dev/kfd", "/dev/dri"}
    with _w.catch_warnings(record=True) as rec:
        _w.simplefilter("always")
        assert get_device_mounts({"bogus/dev": 1}) == []
    assert any("bogus" in str(r.message) for r in rec)


def test_scheduler_factories_direct():
    """Every scheduler module exposes create_scheduler(session_name)."""
    from torchx_amd.schedulers import docker_scheduler, local_scheduler

    s = local_scheduler.create_scheduler("t")
    assert s.backend == "local_cwd"
    d = docker_scheduler.create_scheduler("t")
    assert d.backend == "local_docker"
