"""End-to-end launcher tests on local_cwd: real Popen, real c10d rendezvous
(gloo, multi-process single host) — the reference test strategy's layers (a)
request generation, (b) real local execution, (c) collectives without a
cluster."""

import os
import sys
import tempfile
import time

import pytest

from torchx_amd.runner import get_runner
from torchx_amd.specs import AppState


pytestmark = pytest.mark.subprocess_heavy

def _wait(runner, handle, timeout=120.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        status = runner.status(handle)
        if status is not None and status.is_terminal():
            return status
        time.sleep(0.25)
    raise TimeoutError(f"app {handle} did not finish")


def test_echo_end_to_end(tmp_path):
    with get_runner("test") as runner:
        handle = runner.run_component(
            "utils.echo",
            ["--msg", "hello-mi355x", "--num_replicas", "2"],
            scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        )
        status = _wait(runner, handle)
        assert status.state == AppState.SUCCEEDED
        lines = list(runner.log_lines(handle, "echo", k=0))
        assert any("hello-mi355x" in ln for ln in lines)


def test_sh_failure_is_reported(tmp_path):
    with get_runner("test") as runner:
        handle = runner.run_component(
            "utils.sh", ["sh", "-c", "exit 3"],
            scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        )
        status = _wait(runner, handle)
        assert status.state == AppState.FAILED


def test_touch_and_cancel(tmp_path):
    with get_runner("test") as runner:
        handle = runner.run_component(
            "utils.sh", ["sleep", "60"],
            scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        )
        runner.cancel(handle)
        status = _wait(runner, handle, timeout=30)
        assert status.state in (AppState.CANCELLED, AppState.FAILED)


def test_dryrun_popen_request():
    with get_runner("test") as runner:
        info = runner.dryrun_component(
            "dist.ddp",
            ["--script", "train.py", "--j", "1x2", "--name", "exp/run1"],
            scheduler="local_cwd",
            cfg={"auto_set_hip_visible_devices": False},
        )
        req = info.request
        assert len(req.role_params) == 1
        params = next(iter(req.role_params.values()))
        assert len(params) == 1  # 1 node
        args = params[0].args
        assert args[0] == "python3"
        assert "torchx_amd.agent" in args
        assert "--nproc-per-node" in args
        assert args[args.index("--nproc-per-node") + 1] == "2"
        # env contract
        env = params[0].env
        assert env["TORCHX_RANK0_HOST"] == "localhost"
        assert "TORCHELASTIC_ERROR_FILE" in env
        assert "PET_LOG_DIR" in env


def test_ddp_gloo_world2(tmp_path):
    """BASELINE config 1: dist.ddp -j 1x2 smoke payload over gloo."""
    with get_runner("test") as runner:
        handle = runner.run_component(
            "dist.ddp",
            ["--m", "torchx_amd.apps.compute_world_size", "--j", "1x2"],
            scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        )
        status = _wait(runner, handle, timeout=180)
        lines = list(
            runner.log_lines(handle, status.roles[0].role
                             if status.roles else "compute_world_size")
        )
        assert status.state == AppState.SUCCEEDED, (
            f"{status}\nlogs:\n" + "\n".join(lines[-30:])
        )
        assert any("computed world size = 2" in ln for ln in lines), lines[-30:]
        # agent wrote per-rank stdout files (torchrun --tee layout:
        # <PET_LOG_DIR>/<restart>/<local_rank>/stdout.log)
        import glob

        rank_logs = glob.glob(str(tmp_path) + "/**/0/0/stdout.log",
                              recursive=True)
        assert rank_logs, list(tmp_path.rglob("*.log"))


def test_ddp_elastic_restart(tmp_path):
    """Worker kill -> re-rendezvous -> success (max_restarts=1)."""
    script = tmp_path / "flaky.py"
    marker = tmp_path / "marker"
    script.write_text(
        f"""
import os, sys
sys.path.insert(0, {os.getcwd()!r})
from torchx_amd.apps.compute_world_size import compute_world_size
marker = {str(marker)!r}
if os.environ["RANK"] == "1" and not os.path.exists(marker):
    open(marker, "w").close()
    sys.exit(17)
compute_world_size()
"""
    )
    with get_runner("test") as runner:
        handle = runner.run_component(
            "dist.ddp",
            ["--script", str(script), "--j", "1x2", "--max_retries", "1"],
            scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        )
        status = _wait(runner, handle, timeout=180)
        lines = list(runner.log_lines(handle, status.roles[0].role))
        assert status.state == AppState.SUCCEEDED, (
            f"{status}\nlogs:\n" + "\n".join(lines[-40:])
        )
        assert marker.exists()


def test_runner_list(tmp_path):
    with get_runner("test") as runner:
        handle = runner.run_component(
            "utils.echo", ["--msg", "x"], scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False},
        )
        _wait(runner, handle)
        apps = runner.list("local_cwd")
        assert any(handle.endswith(a.app_id) for a in apps)


def test_launch_latency_tool():
    """BASELINE headline latency metric is measurable end to end."""
    import json
    import subprocess
    import sys as _sys
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    out = subprocess.run(
        [_sys.executable, str(repo / "tools" / "launch_latency.py"),
         "--nproc", "2"],
        capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stdout + out.stderr
    result = json.loads(out.stdout.strip().splitlines()[-1])
    assert result["metric"] == "launch_to_first_step_seconds"
    assert result["value"] and result["value"] > 0


def test_ddp_rocprof_argv():
    """--rocprof reaches the agent argv (SURVEY §5.1 profiling bridge)."""
    from torchx_amd.specs.builders import materialize_appdef
    from torchx_amd.specs.finder import get_component

    comp = get_component("dist.ddp")
    app = materialize_appdef(
        comp.fn, ["-j", "1x2", "--script", "t.py", "--rocprof", "True"],
    )
    assert "--rocprof" in app.roles[0].args


def test_ddp_elastic_nnodes_and_name():
    from torchx_amd.specs.builders import materialize_appdef
    from torchx_amd.specs.finder import get_component

    comp = get_component("dist.ddp")
    app = materialize_appdef(comp.fn, [
        "-j", "2:4x8", "--script", "train.py", "--name", "exp1/run9",
        "--env", "FOO=bar", "--debug", "True",
    ])
    role = app.roles[0]
    assert app.name == "run9"
    assert role.num_replicas == 4          # max nodes
    assert role.min_replicas == 2          # elastic minimum
    assert "--nnodes" in role.args
    assert role.args[role.args.index("--nnodes") + 1] == "2:4"
    assert role.env["FOO"] == "bar"
    assert role.env["NCCL_DESYNC_DEBUG"] == "1"   # debug preset
    # multi-node rendezvous points at the scheduler-provided rank0 host
    ep = role.args[role.args.index("--rdzv-endpoint") + 1]
    assert "TORCHX_RANK0_HOST" in ep


def test_ddp_mounts_parse():
    from torchx_amd.specs import BindMount
    from torchx_amd.specs.builders import materialize_appdef
    from torchx_amd.specs.finder import get_component

    comp = get_component("dist.ddp")
    app = materialize_appdef(comp.fn, [
        "-j", "1x2", "--script", "t.py",
        "--mounts", "type=bind,src=/data,dst=/mnt/data,readonly",
    ])
    m = app.roles[0].mounts[0]
    assert isinstance(m, BindMount)
    assert m.src_path == "/data" and m.dst_path == "/mnt/data"
    assert m.read_only


def test_success_manifest_written(tmp_path):
    import os

    with get_runner("t") as runner:
        handle = runner.run_component(
            "utils.echo", ["--msg", "done"], scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path),
                 "auto_set_hip_visible_devices": False},
        )
        status = _wait(runner, handle, timeout=60)
        assert status.state == AppState.SUCCEEDED
    # the local scheduler writes a SUCCESS manifest on terminal close
    found = []
    for root, _, files in os.walk(tmp_path):
        if "SUCCESS" in files:
            found.append(root)
    assert found, f"no SUCCESS manifest under {tmp_path}"


def test_local_scheduler_lru_eviction(tmp_path):
    from torchx_amd.schedulers.local_scheduler import LocalScheduler
    from torchx_amd.specs import AppDef, Resource, Role

    s = LocalScheduler("t", cache_size=2)
    handles = []
    for i in range(3):
        app = AppDef(name=f"e{i}", roles=[
            Role(name="r", image="/", entrypoint="true", num_replicas=1,
                 resource=Resource(cpu=1, gpu=0, memMB=32)),
        ])
        info = s.submit_dryrun(
            app, {"log_dir": str(tmp_path),
                  "auto_set_hip_visible_devices": False})
        handles.append(s.schedule(info))
    import time

    deadline = time.time() + 30
    while time.time() < deadline:
        live = [s.describe(h) for h in handles]
        if all(d is None or d.state == AppState.SUCCEEDED for d in live):
            break
        time.sleep(0.2)
    # submitting one more evicts the oldest finished app beyond cache_size
    app = AppDef(name="e3", roles=[
        Role(name="r", image="/", entrypoint="true", num_replicas=1,
             resource=Resource(cpu=1, gpu=0, memMB=32)),
    ])
    info = s.submit_dryrun(
        app, {"log_dir": str(tmp_path), "auto_set_hip_visible_devices": False})
    s.schedule(info)
    assert s.describe(handles[0]) is None  # evicted
    s.close()


def test_integ_matrix_all_cheap_components(tmp_path):
    """Reference-style integration matrix: dryrun-print then actually run
    every cheap builtin through local_cwd (parity:
    torchx/components/integration_tests/integ_tests.py)."""
    providers = [
        ("utils.echo", ["--msg", "integ-echo"]),
        ("utils.touch", ["--file", str(tmp_path / "touched.txt")]),
        ("utils.sh", ["echo", "integ-sh"]),
        ("utils.python", ["-c", "print('integ-python')"]),
        ("utils.booth", ["--x1", "1.0", "--x2", "3.0",
                         "--tracker_base", str(tmp_path / "booth")]),
    ]
    with get_runner("integ") as runner:
        for name, args in providers:
            info = runner.dryrun_component(
                name, args, scheduler="local_cwd",
                cfg={"log_dir": str(tmp_path),
                     "auto_set_hip_visible_devices": False},
            )
            assert info.request is not None, name
            handle = runner.run_component(
                name, args, scheduler="local_cwd",
                cfg={"log_dir": str(tmp_path),
                     "auto_set_hip_visible_devices": False},
            )
            status = _wait(runner, handle, timeout=120)
            assert status.state == AppState.SUCCEEDED, (name, status)
    assert (tmp_path / "touched.txt").exists()


def test_multi_role_ps_style_app(tmp_path):
    # parameter-server-style AppDef: three heterogeneous roles in one app
    # (reference parity: cli/test/container ps_main/train_main/reader_main)
    from torchx_amd.specs import AppDef, Resource, Role, macros

    def role(name, num_replicas, msg):
        return Role(
            name=name, image="/", entrypoint=sys.executable,
            args=["-c",
                  f"print('{msg}', '{macros.replica_id}')"],
            num_replicas=num_replicas,
            resource=Resource(cpu=1, gpu=0, memMB=256),
        )

    app = AppDef(name="psjob", roles=[
        role("ps", 1, "ps-up"),
        role("trainer", 2, "train-step"),
        role("reader", 2, "read-batch"),
    ])
    with get_runner("test") as runner:
        handle = runner.run(
            app, scheduler="local_cwd",
            cfg={"log_dir": str(tmp_path),
                 "auto_set_hip_visible_devices": False},
        )
        status = _wait(runner, handle)
        assert status.state == AppState.SUCCEEDED
        # per-role status carries every replica of every role
        by_role = {r.role: r for r in status.roles}
        assert set(by_role) == {"ps", "trainer", "reader"}
        assert len(by_role["trainer"].replicas) == 2
        # logs are addressable per role/replica; ${replica_id} substituted
        trainer1 = list(runner.log_lines(handle, "trainer", k=1))
        assert any("train-step 1" in ln for ln in trainer1)
        reader0 = list(runner.log_lines(handle, "reader", k=0))
        assert any("read-batch 0" in ln for ln in reader0)
