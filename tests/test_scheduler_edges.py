"""Scheduler request-generation edge cases and runner semantics (parity:
the reference's per-scheduler test depth — slurm_scheduler_test.py,
kubernetes_scheduler_test.py, runner/api_test.py)."""

import copy
import json
from unittest.mock import MagicMock, patch

import pytest

from torchx_amd.runner import get_runner
from torchx_amd.specs import (
    AppDef,
    AppState,
    BindMount,
    Resource,
    RetryPolicy,
    Role,
    macros,
)


def _role(**kw):
    base = dict(
        name="trainer", image="img", entrypoint="python",
        args=["-m", "app", "--rank", macros.replica_id],
        env={"E": "1"},
        resource=Resource(cpu=4, gpu=8, memMB=1 << 20),
        num_replicas=2,
    )
    base.update(kw)
    return Role(**base)


class TestSlurmEdges:
    def _req(self, app, cfg=None):
        from torchx_amd.schedulers.slurm_scheduler import SlurmScheduler

        s = SlurmScheduler("test")
        with patch.object(SlurmScheduler, "_slurm_version",
                          return_value=(24, 11), create=True):
            return s.submit_dryrun(app, cfg or {}).request

    def test_hetjob_groups_and_requeue(self):
        app = AppDef(name="train", roles=[
            _role(max_retries=2, retry_policy=RetryPolicy.REPLICA),
        ])
        req = self._req(app)
        script = req.materialize() if hasattr(req, "materialize") else str(req)
        assert "hetjob" in script            # one group per replica
        assert "scontrol requeue" in script  # retry loop
        assert "TORCHX_MAX_RETRIES=2" in script

    def test_macros_resolved_per_replica(self):
        app = AppDef(name="train", roles=[_role()])
        req = self._req(app)
        script = req.materialize() if hasattr(req, "materialize") else str(req)
        # replica ids materialized (0 and 1), app_id deferred to $SLURM_JOB_ID
        assert "--rank 0" in script and "--rank 1" in script

    def test_partition_passthrough(self):
        app = AppDef(name="t", roles=[_role(num_replicas=1)])
        req = self._req(app, {"partition": "gpu-mi355x"})
        script = req.materialize() if hasattr(req, "materialize") else str(req)
        assert "gpu-mi355x" in script


class TestK8sEdges:
    def _resource(self, app, cfg=None):
        from torchx_amd.schedulers.kubernetes_scheduler import (
            KubernetesScheduler,
        )

        s = KubernetesScheduler("test")
        base = {"queue": "default"}
        base.update(cfg or {})
        return s.submit_dryrun(app, base).request.resource

    def test_retry_policy_mapping(self):
        res = self._resource(AppDef(name="a", roles=[
            _role(max_retries=3, retry_policy=RetryPolicy.APPLICATION),
        ]))
        task = res["spec"]["tasks"][0]
        assert task["maxRetry"] == 3
        events = {p["event"] for p in task["policies"]}
        assert "PodEvicted" in events  # APPLICATION -> RestartJob

    def test_device_mount_privileged(self):
        from torchx_amd.specs import DeviceMount

        role = _role(mounts=[DeviceMount(src_path="/dev/kfd",
                                         dst_path="/dev/kfd",
                                         permissions="rwm")],
                     num_replicas=1)
        res = self._resource(AppDef(name="a", roles=[role]))
        pod = res["spec"]["tasks"][0]["template"]
        c = pod["spec"]["containers"][0]
        assert c["securityContext"]["privileged"]

    def test_gpu_resource_is_amd(self):
        res = self._resource(AppDef(name="a", roles=[_role(num_replicas=1)]))
        limits = (res["spec"]["tasks"][0]["template"]["spec"]["containers"]
                  [0]["resources"]["limits"])
        assert limits["amd.com/gpu"] == 8
        assert "nvidia.com/gpu" not in limits

    def test_shm_volume_for_rccl(self):
        res = self._resource(AppDef(name="a", roles=[_role(num_replicas=1)]))
        spec = res["spec"]["tasks"][0]["template"]["spec"]
        shm = [v for v in spec["volumes"] if v["name"] == "dshm"]
        assert shm and shm[0]["emptyDir"]["medium"] == "Memory"

    def test_rank0_env_wiring(self):
        res = self._resource(AppDef(name="a", roles=[_role()]))
        envs0 = {e["name"]: e["value"]
                 for e in (res["spec"]["tasks"][0]["template"]["spec"]
                           ["containers"][0]["env"])}
        assert envs0.get("TORCHX_RANK0_HOST") == "localhost"
        # non-rank0 replicas rely on VC_..._HOSTS via rank0_env macro
        envs1 = {e["name"]: e["value"]
                 for e in (res["spec"]["tasks"][1]["template"]["spec"]
                           ["containers"][0]["env"])}
        assert "TORCHX_RANK0_HOST" not in envs1


class TestRunnerSemantics:
    def test_dryrun_does_not_mutate_caller_appdef(self):
        app = AppDef(name="a", roles=[_role(num_replicas=1)])
        before = copy.deepcopy(app)
        with get_runner("t") as runner:
            runner.dryrun(app, "local_cwd",
                          cfg={"auto_set_hip_visible_devices": False})
        assert app.roles[0].env == before.roles[0].env
        assert app.roles[0].args == before.roles[0].args

    def test_dryrun_injects_session_env(self):
        app = AppDef(name="a", roles=[_role(num_replicas=1)])
        with get_runner("sess-x") as runner:
            info = runner.dryrun(app, "local_cwd",
                                 cfg={"auto_set_hip_visible_devices": False})
        role = info._app.roles[0]
        assert "TORCHX_JOB_ID" in role.env
        assert role.env.get("TORCHX_INTERNAL_SESSION_ID")

    def test_job_id_reaches_materialized_request(self):
        # regression: TORCHX_JOB_ID must be injected
        # BEFORE submit_dryrun so the ${app_id} macro is substituted into
        # the materialized PopenRequest replica env — otherwise every real
        # job sees <unset_run_id> in AppRun.run_from_env().
        app = AppDef(name="a", roles=[_role(num_replicas=2)])
        with get_runner("sess-y") as runner:
            info = runner.dryrun(app, "local_cwd",
                                 cfg={"auto_set_hip_visible_devices": False})
        for params in info.request.role_params.values():
            for p in params:
                job_id = p.env.get("TORCHX_JOB_ID")
                assert job_id is not None
                assert "${app_id}" not in job_id
                # handle embeds the scheduler-assigned app id
                assert job_id.startswith("local_cwd://sess-y/a-")
                assert p.env.get("TORCHX_INTERNAL_SESSION_ID")

    def test_run_rejects_unknown_scheduler(self):
        with get_runner("t") as runner:
            with pytest.raises(Exception):
                runner.dryrun(AppDef(name="a", roles=[_role()]), "nope")

    def test_parent_run_id_env(self, monkeypatch):
        monkeypatch.setenv("TORCHX_PARENT_RUN_ID", "local_cwd://s/parent1")
        app = AppDef(name="a", roles=[_role(num_replicas=1)])
        with get_runner("t") as runner:
            info = runner.dryrun(app, "local_cwd",
                                 cfg={"auto_set_hip_visible_devices": False})
        assert (info._app.roles[0].env.get("TORCHX_PARENT_RUN_ID")
                == "local_cwd://s/parent1")


class TestNumaAffinity:
    def _fake_sysfs(self, tmp_path, cards):
        # cards: list of (render_minor, vendor, numa_node)
        for minor, vendor, numa in cards:
            d = tmp_path / "class" / "drm" / f"renderD{minor}" / "device"
            d.mkdir(parents=True)
            (d / "vendor").write_text(vendor + "\n")
            (d / "numa_node").write_text(str(numa) + "\n")
        return str(tmp_path)

    def test_numa_map_from_sysfs(self, tmp_path):
        from torchx_amd.schedulers.devices import _read_device_numa_map

        root = self._fake_sysfs(tmp_path, [
            (128, "0x1002", 0), (129, "0x1002", 0),
            (130, "0x1002", 1), (131, "0x1002", 1),
            (132, "0x10de", 7),  # non-AMD card is skipped entirely
        ])
        assert _read_device_numa_map(root) == {0: 0, 1: 0, 2: 1, 3: 1}

    def test_numa_bind_args(self):
        from torchx_amd.schedulers.devices import numa_bind_args

        nmap = {0: 0, 1: 0, 2: 1, 3: 1}
        assert numa_bind_args("0,1", nmap, numactl="/usr/bin/numactl") == [
            "/usr/bin/numactl", "--cpunodebind=0", "--membind=0",
        ]
        assert numa_bind_args("2", nmap, numactl="/usr/bin/numactl") == [
            "/usr/bin/numactl", "--cpunodebind=1", "--membind=1",
        ]
        # spans two nodes -> no binding
        assert numa_bind_args("1,2", nmap, numactl="/usr/bin/numactl") == []
        # unknown node -> no binding
        assert numa_bind_args("9", nmap, numactl="/usr/bin/numactl") == []
        # numactl missing -> no binding
        assert numa_bind_args("0", nmap, numactl="") == []
        assert numa_bind_args(None, nmap, numactl="/usr/bin/numactl") == []

    def test_replica_args_carry_numactl_prefix(self, monkeypatch):
        from torchx_amd.schedulers import devices as dev_mod
        from torchx_amd.schedulers.local_scheduler import LocalScheduler
        from torchx_amd.specs import AppDef, Resource

        monkeypatch.setattr(dev_mod, "hip_device_count", lambda: 8)
        monkeypatch.setattr(
            "torchx_amd.schedulers.local_scheduler.numa_bind_args",
            lambda devs: ["/usr/bin/numactl", "--cpunodebind=0",
                          "--membind=0"] if devs == "0,1,2,3" else
                         ["/usr/bin/numactl", "--cpunodebind=1",
                          "--membind=1"],
        )
        monkeypatch.setattr(
            "torchx_amd.schedulers.local_scheduler.partition_devices",
            lambda rr, gg: {"trainer": ["0,1,2,3", "4,5,6,7"]},
        )
        s = LocalScheduler("t")
        role = _role(num_replicas=2,
                     resource=Resource(cpu=8, gpu=4, memMB=1024))
        app = AppDef(name="x", roles=[role])
        info = s._submit_dryrun(app, {"auto_set_hip_visible_devices": True,
                                      "numa_affinity": True})
        p0, p1 = info.request.role_params["trainer"]
        assert p0.args[:3] == ["/usr/bin/numactl", "--cpunodebind=0",
                               "--membind=0"]
        assert p1.args[:3] == ["/usr/bin/numactl", "--cpunodebind=1",
                               "--membind=1"]
        assert p0.env["HIP_VISIBLE_DEVICES"] == "0,1,2,3"
        # opt off -> clean argv
        info2 = s._submit_dryrun(app, {"auto_set_hip_visible_devices": True,
                                       "numa_affinity": False})
        assert "numactl" not in info2.request.role_params["trainer"][0].args[0]


class TestLocalSchedulerEdges:
    def test_bind_mount_rejected(self, tmp_path):
        # local_cwd has no mount support (compat matrix: mounts x)
        from torchx_amd.schedulers.local_scheduler import LocalScheduler

        s = LocalScheduler("t")
        role = _role(num_replicas=1,
                     mounts=[BindMount(src_path="/a", dst_path="/b")])
        app = AppDef(name="x", roles=[role])
        with pytest.raises(Exception):
            s.submit_dryrun(app, {"auto_set_hip_visible_devices": False})

    def test_hip_device_partitioning_wiring(self, tmp_path):
        from torchx_amd.schedulers.local_scheduler import LocalScheduler

        s = LocalScheduler("t")
        app = AppDef(name="x", roles=[
            Role(name="w", image="/", entrypoint="echo", num_replicas=2,
                 resource=Resource(cpu=1, gpu=2, memMB=64)),
        ])
        with patch("torchx_amd.schedulers.devices.hip_device_count",
                   return_value=4):
            info = s.submit_dryrun(
                app, {"auto_set_hip_visible_devices": True,
                      "log_dir": str(tmp_path)})
        params = next(iter(info.request.role_params.values()))
        assert params[0].env.get("HIP_VISIBLE_DEVICES") == "0,1"
        assert params[1].env.get("HIP_VISIBLE_DEVICES") == "2,3"


class TestRegistryAndRunopts:
    def test_all_four_builtin_schedulers(self):
        from torchx_amd.schedulers import get_scheduler_factories

        factories = get_scheduler_factories()
        for name in ("local_cwd", "local_docker", "slurm", "kubernetes"):
            assert name in factories, factories

    def test_default_scheduler_is_first(self):
        from torchx_amd.schedulers import get_default_scheduler_name

        assert get_default_scheduler_name() == "local_cwd"

    def test_every_scheduler_has_runopts_help(self):
        from torchx_amd.runner import get_runner

        with get_runner("t") as runner:
            for name in ("local_cwd", "local_docker", "slurm", "kubernetes"):
                opts = runner.scheduler_run_opts(name)
                assert opts is not None
                # every option carries help text
                for oname, opt in getattr(opts, "_opts", {}).items():
                    assert opt.help, f"{name}.{oname} has no help"


class TestSlurmList:
    def test_list_merges_hetjob_groups(self):
        from unittest.mock import patch as _patch

        from torchx_amd.schedulers.slurm_scheduler import SlurmScheduler

        payload = {"jobs": [
            {"job_id": 300, "het_job_id": 300, "het_job_offset": 0,
             "name": "t-0", "job_state": ["RUNNING"]},
            {"job_id": 301, "het_job_id": 300, "het_job_offset": 1,
             "name": "t-1", "job_state": ["RUNNING"]},
        ]}
        import json as _json

        with _patch("subprocess.check_output",
                    return_value=_json.dumps(payload).encode()):
            apps = SlurmScheduler("t").list()
        # two het groups of one job -> ONE listed app
        assert len(apps) == 1
        assert apps[0].app_id == "300"
