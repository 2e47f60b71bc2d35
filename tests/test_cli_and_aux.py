"""CLI, config, tracker, events, plugins, finder tests."""

import json
import os

import pytest

from torchx_amd.cli.main import create_parser, main as cli_main
from torchx_amd.runner import config as torchx_config
from torchx_amd.specs.finder import (
    ComponentNotFoundException,
    get_builtin_source,
    get_component,
    get_components,
)


def test_builtins_discovery():
    comps = get_components()
    for expected in ("dist.ddp", "dist.spmd", "utils.echo", "utils.sh",
                     "utils.python", "utils.binary", "utils.copy",
                     "utils.touch", "utils.booth"):
        assert expected in comps, f"missing builtin {expected}"


def test_get_component_custom_file(tmp_path):
    f = tmp_path / "comp.py"
    f.write_text(
        "from torchx_amd.specs import AppDef, Role\n"
        "def my(name: str = 'x') -> AppDef:\n"
        "    return AppDef(name=name, roles=[Role(name='r', image='/')])\n"
    )
    comp = get_component(f"{f}:my")
    assert comp.fn_name == "my"
    with pytest.raises(ComponentNotFoundException):
        get_component(f"{f}:nope")
    with pytest.raises(ComponentNotFoundException):
        get_component("no.such.component")


def test_builtin_source():
    src = get_builtin_source("utils.echo")
    assert "def echo" in src


def test_cli_builtins(capsys):
    assert cli_main(["builtins"]) == 0
    out = capsys.readouterr().out
    assert "dist.ddp" in out


def test_cli_runopts(capsys):
    assert cli_main(["runopts", "local_cwd"]) == 0
    out = capsys.readouterr().out
    assert "log_dir" in out


def test_cli_run_dryrun(capsys):
    rc = cli_main([
        "run", "--dryrun", "-s", "local_cwd",
        "utils.echo", "--msg", "hi",
    ])
    assert rc == 0
    out = capsys.readouterr().out
    assert "SCHEDULER REQUEST" in out


def test_cli_run_end_to_end(tmp_path, capsys):
    rc = cli_main([
        "run", "-s", "local_cwd",
        "-cfg", f"log_dir={tmp_path},auto_set_hip_visible_devices=false",
        "utils.echo", "--msg", "cli-e2e",
    ])
    assert rc == 0


def test_torchxconfig(tmp_path, monkeypatch):
    cfg = tmp_path / ".torchxconfig"
    cfg.write_text(
        "[local_cwd]\nlog_dir = /tmp/xyz\n\n"
        "[component:dist.ddp]\nj = 2x8\n\n"
        "[cli:run]\ncomponent = utils.echo\n\n"
        "[torchx:tracker]\nfsspec = /tmp/tracker-root\n"
    )
    monkeypatch.setenv("TORCHXCONFIG", str(cfg))
    c = {}
    torchx_config.load("local_cwd", c)
    assert c["log_dir"] == "/tmp/xyz"
    sections = torchx_config.load_sections("component")
    assert sections["dist.ddp"]["j"] == "2x8"
    assert torchx_config.get_config("cli", "run", "component") == "utils.echo"
    trackers = torchx_config.get_configured_trackers()
    assert trackers == {"fsspec": "/tmp/tracker-root"}


def test_tracker_fsspec(tmp_path):
    from torchx_amd.tracker.fsspec import FsspecTracker

    t = FsspecTracker(str(tmp_path))
    t.add_metadata("run1", lr=0.1, name="exp")
    t.add_metadata("run1", lr=0.2)
    assert t.metadata("run1") == {"lr": 0.2, "name": "exp"}
    t.add_artifact("run1", "ckpt", "/tmp/ckpt.pt")
    assert t.artifacts("run1")["ckpt"] == "/tmp/ckpt.pt"
    t.add_source("run1", "run0")
    assert list(t.sources("run1")) == ["run0"]
    assert set(t.run_ids()) == {"run1", "run0"}


def test_app_run_from_env(tmp_path):
    from torchx_amd.tracker.api import AppRun

    env = {
        "TORCHX_JOB_ID": "local_cwd://s/app1",
        "TORCHX_TRACKERS": "fsspec",
        "TORCHX_TRACKER_FSSPEC_CONFIG": str(tmp_path),
        "TORCHX_PARENT_RUN_ID": "local_cwd://s/app0",
    }
    run = AppRun.run_from_env(env)
    run.add_metadata(step=1)
    from torchx_amd.tracker.fsspec import FsspecTracker

    t = FsspecTracker(str(tmp_path))
    assert t.metadata("local_cwd://s/app1")["step"] == 1
    assert list(t.sources("local_cwd://s/app1")) == ["local_cwd://s/app0"]


def test_events_log_event():
    from torchx_amd.runner.events import TorchxEvent, log_event

    with log_event("run", "local_cwd", "sess") as ctx:
        pass
    assert ctx.event.wall_time_usec is not None
    try:
        with log_event("fail", "local_cwd", "sess") as ctx2:
            raise ValueError("boom")
    except ValueError:
        pass
    assert ctx2.event.exception_type == "ValueError"
    assert "boom" in ctx2.event.exception_message
    assert json.loads(ctx2.event.serialize())["api"] == "fail"


def test_plugins_register():
    from torchx_amd.plugins import PluginType, register, registry
    from torchx_amd.specs import Resource, resource

    @register.named_resource("test.gpu8", powers_of_two_gpus=True)
    def _res():
        return Resource(cpu=64, gpu=8, memMB=1 << 20)

    assert resource(h="test.gpu8").gpu == 8
    assert resource(h="test.gpu8_4").gpu == 4
    assert resource(h="test.gpu8_1").gpu == 1

    @register.scheduler("test_sched")
    def _sched_factory(session_name: str, **kw):
        return None

    assert "test_sched" in registry().scheduler_factories()


def test_result_tracker(tmp_path):
    from torchx_amd.runtime.tracking import FsspecResultTracker

    t = FsspecResultTracker(str(tmp_path))
    t[0] = {"metric": 42.0}
    assert t[0] == {"metric": 42.0}


def test_devices_partition():
    from torchx_amd.schedulers.devices import device_env, partition_devices

    assign = partition_devices(
        {"trainer": 2, "ps": 1}, {"trainer": 4, "ps": 0}, total_gpus=8
    )
    assert assign["trainer"] == ["0,1,2,3", "4,5,6,7"]
    assert assign["ps"] == [None]
    env = device_env("0,1")
    assert env["HIP_VISIBLE_DEVICES"] == "0,1"
    assert env["ROCR_VISIBLE_DEVICES"] == "0,1"
    # oversubscription -> no pinning
    assign2 = partition_devices({"t": 4}, {"t": 4}, total_gpus=8)
    assert assign2["t"] == [None] * 4


def test_cli_run_stdin_json(tmp_path):
    import json as _json
    import subprocess
    import sys as _sys

    spec = {"component": "utils.echo",
            "component_args": ["--msg", "stdin-mode"],
            "scheduler": "local_cwd", "dryrun": True}
    out = subprocess.run(
        [_sys.executable, "-m", "torchx_amd.cli.main", "run", "--stdin"],
        input=_json.dumps(spec), capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr
    assert "stdin-mode" in out.stdout


def test_cli_status_describe_list_cancel(tmp_path, capsys):
    # local_cwd state is per-session (as in the reference): a run handle
    # resolves within the owning Runner; a fresh session reports not-found
    rc = cli_main([
        "run", "-s", "local_cwd",
        "-cfg", f"log_dir={tmp_path},auto_set_hip_visible_devices=false",
        "utils.echo", "--msg", "cycle",
    ])
    assert rc == 0
    out = capsys.readouterr().out
    handle = out.splitlines()[0].strip()
    assert handle.startswith("local_cwd://")
    assert cli_main(["list", "-s", "local_cwd"]) == 0
    # unknown-app error paths exit non-zero, don't crash
    assert cli_main(["status", handle]) == 1
    assert cli_main(["describe", handle]) == 1


def test_cli_configure_dumps_template(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    assert cli_main(["configure", "-s", "local_cwd"]) == 0
    cfg = tmp_path / ".torchxconfig"
    assert cfg.exists()
    text = cfg.read_text()
    assert "[local_cwd]" in text
    # generated from the scheduler's ACTUAL runopts, not a static template
    assert "log_dir" in text
    assert "auto_set_hip_visible_devices" in text
    assert "[kubernetes]" not in text  # -s filter respected


def test_configure_all_schedulers_marks_required(tmp_path):
    from torchx_amd.runner import config as torchx_config

    path = str(tmp_path / "cfg.ini")
    torchx_config.dump(path)
    text = open(path).read()
    # every registered scheduler gets a section
    for section in ("[local_cwd]", "[local_docker]", "[slurm]",
                    "[kubernetes]"):
        assert section in text, section
    # required opts surface as uncommented #FIXME lines
    assert "queue = #FIXME" in text
    # optional opts are commented with their default
    assert "# namespace = default" in text


def test_configure_unknown_scheduler_raises(tmp_path):
    import pytest

    from torchx_amd.runner import config as torchx_config

    with pytest.raises(ValueError):
        torchx_config.dump(str(tmp_path / "x.ini"), schedulers=["nope"])


def test_cli_tracker_cmds(tmp_path, monkeypatch):
    from torchx_amd.tracker.fsspec import FsspecTracker

    t = FsspecTracker(str(tmp_path))
    t.add_metadata("run7", lr=0.5)
    t.add_artifact("run7", "ckpt", "/x/ckpt.pt")
    cfg = tmp_path / ".torchxconfig"
    cfg.write_text(f"[torchx:tracker]\nfsspec = {tmp_path}\n")
    monkeypatch.setenv("TORCHXCONFIG", str(cfg))
    assert cli_main(["tracker", "list", "jobs"]) == 0
    assert cli_main(["tracker", "list", "metadata", "--run_id", "run7"]) == 0


def test_cli_entry_point_subcommands(monkeypatch):
    # custom subcommand via the torchx_amd.cli.cmds entry-point group;
    # an entry point named like a builtin ("status") overrides it
    import importlib.metadata as md

    from torchx_amd.cli import main as cli_main

    calls = []

    class HelloCmd:
        """say hello"""

        def add_arguments(self, parser):
            parser.add_argument("--who", default="world")

        def run(self, args):
            calls.append(("hello", args.who))
            return 0

    class StatusOverride:
        def add_arguments(self, parser):
            parser.add_argument("extra")

        def run(self, args):
            calls.append(("status", args.extra))
            return 0

    class FakeEp:
        def __init__(self, name, obj):
            self.name = name
            self._obj = obj

        def load(self):
            return self._obj

    class BrokenEp:
        name = "broken"

        def load(self):
            raise RuntimeError("cli-plugin-boom")

    class FakeEps:
        def select(self, group):
            if group == "torchx_amd.cli.cmds":
                return [FakeEp("hello", HelloCmd),
                        FakeEp("status", StatusOverride),
                        BrokenEp()]
            return []

    monkeypatch.setattr(md, "entry_points", lambda: FakeEps())
    assert cli_main.main(["hello", "--who", "mi355x"]) == 0
    assert cli_main.main(["status", "anything"]) == 0
    assert calls == [("hello", "mi355x"), ("status", "anything")]


def test_cli_run_scheduler_from_config(tmp_path, monkeypatch, capsys):
    # [cli:run] scheduler= supplies the default; -s still wins
    monkeypatch.chdir(tmp_path)
    monkeypatch.delenv("TORCHXCONFIG", raising=False)
    (tmp_path / ".torchxconfig").write_text(
        "[cli:run]\nscheduler = slurm\n"
    )
    assert cli_main(["run", "--dryrun", "utils.echo", "--msg", "hi"]) == 0
    out = capsys.readouterr().out
    assert "sbatch" in out or "srun" in out  # slurm request generated
