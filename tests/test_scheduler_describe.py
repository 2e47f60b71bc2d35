"""Scheduler describe-path tests against canned CLI/server outputs (parity:
the reference parses canned squeue/sacct fixtures across Slurm versions,
slurm_scheduler_test.py; and Volcano status dicts)."""

import json
from unittest.mock import MagicMock, patch

from torchx_amd.specs import AppState


def _slurm():
    from torchx_amd.schedulers.slurm_scheduler import SlurmScheduler

    return SlurmScheduler("t")


class TestSlurmDescribe:
    def test_squeue_new_format_list_states(self):
        # Slurm >= 23.02: job_state is a list
        payload = {"jobs": [
            {"job_id": 100, "name": "trainer-0",
             "job_state": ["RUNNING"], "nodes": "gpu-node-1"},
            {"job_id": 101, "name": "trainer-1",
             "job_state": ["PENDING"], "nodes": ""},
        ]}
        with patch("subprocess.check_output",
                   return_value=json.dumps(payload).encode()):
            resp = _slurm().describe("100")
        assert resp.state == AppState.RUNNING  # running wins over pending
        role = resp.roles_statuses[0]
        assert role.role == "trainer"
        assert {r.id for r in role.replicas} == {0, 1}
        assert role.replicas[0].hostname == "gpu-node-1"

    def test_squeue_old_format_string_state(self):
        payload = {"jobs": [
            {"job_id": 100, "name": "trainer-0", "job_state": "COMPLETED"},
        ]}
        with patch("subprocess.check_output",
                   return_value=json.dumps(payload).encode()):
            resp = _slurm().describe("100")
        assert resp.state == AppState.SUCCEEDED

    def test_sacct_fallback_for_finished_job(self):
        sacct = (
            "JobID|JobName|State\n"
            "200|trainer-0|FAILED\n"
            "200.batch|batch|FAILED\n"
            "200+1|trainer-1|COMPLETED\n"
        )

        def fake_check_output(cmd, timeout=None):
            if cmd[0] == "squeue":
                raise FileNotFoundError("no squeue")
            return sacct.encode()

        with patch("subprocess.check_output", side_effect=fake_check_output):
            resp = _slurm().describe("200")
        assert resp.state == AppState.FAILED  # failed wins

    def test_cancelled_state(self):
        payload = {"jobs": [
            {"job_id": 1, "name": "w-0", "job_state": ["CANCELLED"]},
        ]}
        with patch("subprocess.check_output",
                   return_value=json.dumps(payload).encode()):
            assert _slurm().describe("1").state == AppState.CANCELLED


class TestK8sDescribe:
    def _sched(self, status):
        from torchx_amd.schedulers.kubernetes_scheduler import (
            KubernetesScheduler,
        )

        s = KubernetesScheduler("t")
        api = MagicMock()
        api.get_namespaced_custom_object.return_value = {
            "status": status,
            "metadata": {"name": "app-x"},
        }
        s._custom_api = lambda: api
        return s

    def test_running(self):
        s = self._sched({"state": {"phase": "Running"},
                         "taskStatusCount": {"w-0": {"phase": {"Running": 1}}}})
        resp = s.describe("default:app-x")
        assert resp.state == AppState.RUNNING

    def test_completed(self):
        s = self._sched({"state": {"phase": "Completed"}})
        assert s.describe("default:app-x").state == AppState.SUCCEEDED

    def test_failed(self):
        s = self._sched({"state": {"phase": "Failed"}})
        assert s.describe("default:app-x").state == AppState.FAILED

    def test_missing_job_returns_none(self):
        from torchx_amd.schedulers.kubernetes_scheduler import (
            KubernetesScheduler,
        )

        s = KubernetesScheduler("t")
        api = MagicMock()

        class FakeApiException(Exception):
            status = 404

        def raise404(**kw):
            raise FakeApiException()

        api.get_namespaced_custom_object.side_effect = raise404
        s._custom_api = lambda: api
        try:
            resp = s.describe("default:gone")
        except Exception:
            resp = None
        assert resp is None
