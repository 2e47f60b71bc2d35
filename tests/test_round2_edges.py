"""Round-2 behavioral edges: linter validators, help formatter, runner
factory env passthrough, Workspace spec strings, CPU parity of the new
fused ops, since/until warning."""

import argparse
import warnings

import pytest
import torch

from torchx_amd import ops
from torchx_amd.specs import AppDef, Resource, Role, Workspace
from torchx_amd.specs.file_linter import ComponentHelpFormatter, validate


class TestLinterValidators:
    def _lint(self, tmp_path, body):
        f = tmp_path / "comp.py"
        f.write_text(
            "from torchx_amd.specs import AppDef\n"
            "from typing import Annotated, Dict, List, Optional, Tuple\n"
            + body
        )
        return validate(str(f), "comp")

    def test_tuple_of_primitives_ok(self, tmp_path):
        msgs = self._lint(tmp_path, (
            "def comp(hw: Tuple[int, int]) -> AppDef:\n"
            "    '''c\n\n    Args:\n        hw: size\n    '''\n"
        ))
        assert not [m for m in msgs if m.severity == "error"]

    def test_single_element_tuple_rejected(self, tmp_path):
        msgs = self._lint(tmp_path, (
            "def comp(hw: Tuple[int]) -> AppDef:\n"
            "    '''c\n\n    Args:\n        hw: size\n    '''\n"
        ))
        assert [m for m in msgs if m.severity == "error"]

    def test_dict_needs_two_params(self, tmp_path):
        msgs = self._lint(tmp_path, (
            "def comp(env: Dict[str]) -> AppDef:\n"
            "    '''c\n\n    Args:\n        env: e\n    '''\n"
        ))
        assert [m for m in msgs if m.severity == "error"]

    def test_annotated_unwraps(self, tmp_path):
        msgs = self._lint(tmp_path, (
            "def comp(j: Annotated[str, '-j'] = '1x1') -> AppDef:\n"
            "    '''c\n\n    Args:\n        j: nodes\n    '''\n"
        ))
        assert not [m for m in msgs if m.severity == "error"]


class TestHelpFormatter:
    def test_required_and_default_suffixes(self):
        p = argparse.ArgumentParser(prog="x",
                                    formatter_class=ComponentHelpFormatter)
        p.add_argument("--name", required=True, help="the name")
        p.add_argument("--retries", default=3, help="retry count")
        text = p.format_help()
        assert "the name (required)" in text
        assert "retry count (default: 3)" in text


class TestRunnerFactoryEnv:
    def test_env_param_passthrough(self, monkeypatch):
        # TORCHX_<SCHED>_<PARAM> env vars become scheduler factory kwargs
        # (reference runner/api.py:131)
        from torchx_amd.runner import Runner

        seen = {}

        class FakeSched:
            def __init__(self):
                self.backend = "fake"

            def close(self):
                pass

        def factory(session_name, **kw):
            seen.update(kw, session=session_name)
            return FakeSched()

        monkeypatch.setenv("TORCHX_FAKE_CACHE_SIZE", "7")
        r = Runner("sess", scheduler_factories={"fake": factory})
        r._scheduler("fake")
        assert seen["cache_size"] == "7"
        assert seen["session"] == "sess"


class TestWorkspaceSpec:
    def test_from_str_variants(self):
        assert not Workspace.from_str(None)
        ws = Workspace.from_str("/a")
        assert ws.is_unmapped_single_project()
        ws2 = Workspace.from_str("/a:dst1,/b:dst2")
        assert ws2.projects == {"/a": "dst1", "/b": "dst2"}
        assert not ws2.is_unmapped_single_project()
        # passthrough of an existing Workspace
        assert Workspace.from_str(ws2) is ws2


class TestFusedOpsCpuParity:
    def test_fused_linear_ce_cpu_matches_torch(self):
        torch.manual_seed(0)
        x = torch.randn(12, 32, requires_grad=True)
        w = torch.randn(64, 32, requires_grad=True)
        t = torch.randint(0, 64, (12,))
        loss = ops.fused_linear_cross_entropy(x, w, t, chunk=8)
        ref = torch.nn.functional.cross_entropy(x @ w.t(), t)
        assert torch.allclose(loss, ref, atol=1e-5)
        loss.backward()
        assert x.grad is not None and w.grad is not None

    def test_fast_linear_cpu_is_linear(self):
        x = torch.randn(4, 16)
        w = torch.randn(8, 16)
        assert torch.allclose(ops.fast_linear(x, w), x @ w.t(), atol=1e-6)


class TestLogIterSinceUntil:
    def test_since_until_warns_and_returns_all(self, tmp_path, caplog):
        import datetime
        import logging

        from torchx_amd.schedulers.local_scheduler import LocalScheduler

        s = LocalScheduler("t")
        role = Role(name="r", image="i", entrypoint="bash",
                    args=["-c", "echo one; echo two"],
                    resource=Resource(cpu=1, gpu=0, memMB=64))
        app = AppDef(name="a", roles=[role])
        cfg = {"log_dir": str(tmp_path),
               "auto_set_hip_visible_devices": False}
        app_id = s.schedule(s.submit_dryrun(app, cfg))
        import time

        deadline = time.time() + 30
        from torchx_amd.specs import is_terminal

        while not is_terminal(s.describe(app_id).state):
            assert time.time() < deadline
            time.sleep(0.1)
        with caplog.at_level(logging.WARNING):
            lines = list(s.log_iter(
                app_id, "r", 0,
                since=datetime.datetime.now(),
            ))
        assert any("ignored" in r.message for r in caplog.records)
        assert any("one" in ln for ln in lines)
        s.close()


class TestWorkerNumaPrefix:
    def test_maps_local_rank_through_visible_devices(self, monkeypatch):
        from torchx_amd.agent.__main__ import worker_numa_prefix
        from torchx_amd.schedulers import devices as dev_mod

        calls = []

        def fake_bind(dev):
            calls.append(dev)
            return ["/usr/bin/numactl", "--cpunodebind=1", "--membind=1"]

        monkeypatch.setattr(dev_mod, "numa_bind_args", fake_bind)
        # replica owns global devices 4..7; worker 2 -> global 6
        pref = worker_numa_prefix(2, {"HIP_VISIBLE_DEVICES": "4,5,6,7"})
        assert calls == ["6"]
        assert pref[0].endswith("numactl")
        # no visible restriction: worker lr maps to global lr
        calls.clear()
        worker_numa_prefix(3, {})
        assert calls == ["3"]
        # out-of-range worker -> no binding
        assert worker_numa_prefix(9, {"HIP_VISIBLE_DEVICES": "0"}) == []

    def test_worker_cpu_affinity_cpulist(self, monkeypatch, tmp_path):
        from torchx_amd.agent.__main__ import worker_cpu_affinity
        from torchx_amd.schedulers import devices as dev_mod

        node = tmp_path / "devices" / "system" / "node" / "node1"
        node.mkdir(parents=True)
        (node / "cpulist").write_text("32-63,160-191\n")
        monkeypatch.setattr(dev_mod, "numa_node_of",
                            lambda dev, numa_map=None: 1 if dev == "6" else -1)
        monkeypatch.setattr(
            dev_mod, "numa_cpulist",
            lambda n, sysfs="/sys": dev_mod.numa_cpulist.__wrapped__(n)
            if False else ("32-63,160-191" if n == 1 else ""))
        aff = worker_cpu_affinity(2, {"HIP_VISIBLE_DEVICES": "4,5,6,7"})
        assert aff == "32-63,160-191"
        assert worker_cpu_affinity(0, {"HIP_VISIBLE_DEVICES": "0"}) == ""
        # parse_cpulist round-trip
        assert dev_mod.parse_cpulist("0-3,8,10-11") == {0, 1, 2, 3, 8, 10, 11}
