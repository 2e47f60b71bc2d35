"""Unit tests for the EP all-to-all helpers (gloo world 2, autograd
round-trip) and the c10d rendezvous module."""

import os
import subprocess

import pytest
import sys
from pathlib import Path

pytestmark = pytest.mark.subprocess_heavy

REPO = Path(__file__).resolve().parent.parent

EP_WORKER = r"""
import os, sys
sys.path.insert(0, %(repo)r)
import torch
import torch.distributed as dist
from torchx_amd.parallel.ep import exchange_counts, expert_all_to_all

dist.init_process_group("gloo")
rank = dist.get_rank()

# rank0 sends [1 row to r0, 2 rows to r1]; rank1 sends [3 to r0, 1 to r1]
send = torch.tensor([1, 2] if rank == 0 else [3, 1])
recv = exchange_counts(send)
assert recv.tolist() == ([1, 3] if rank == 0 else [2, 1]), recv

n = int(send.sum())
x = torch.arange(n * 4, dtype=torch.float32).reshape(n, 4) + 100 * rank
x.requires_grad_(True)
out = expert_all_to_all(x, recv.tolist(), send.tolist(), None)
assert out.shape[0] == int(recv.sum())

# round trip back restores the original rows
back = expert_all_to_all(out, send.tolist(), recv.tolist(), None)
assert torch.equal(back, x.detach()), (back, x)

# autograd: grad of identity-composed a2a is identity
out.backward(out.detach() * 0 + 1.0)
assert torch.equal(x.grad, torch.ones_like(x)), x.grad
if rank == 0:
    print("EP_OK", flush=True)
dist.destroy_process_group()
"""


def test_ep_all_to_all_gloo(tmp_path):
    script = tmp_path / "w.py"
    script.write_text(EP_WORKER % {"repo": str(REPO)})
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes", "1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "0", "--standalone", str(script)],
        env=env, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "EP_OK" in out.stdout + out.stderr


def test_rendezvous_single_node():
    from torchx_amd.agent.rendezvous import C10dRendezvous, free_port

    port = free_port()
    rdzv = C10dRendezvous(f"127.0.0.1:{port}", "unit", 1, 1)
    res = rdzv.join(0)
    assert res.node_rank == 0
    assert res.num_nodes == 1
    assert res.master_port
    rdzv._store = None  # drop the TCPStore


def test_rendezvous_late_joiner_scale_up():
    # a node joining just after round close must not
    # crash. With room below max_nodes it signals a scale-up
    # re-rendezvous and both nodes re-form at the next round.
    import threading

    from torchx_amd.agent.rendezvous import C10dRendezvous, free_port

    port = free_port()
    a = C10dRendezvous(f"127.0.0.1:{port}", "unit-su", 1, 2,
                       timeout=30.0, last_call_timeout=0.2)
    b = C10dRendezvous(f"127.0.0.1:{port}", "unit-su", 1, 2,
                       timeout=30.0, last_call_timeout=0.2)

    results = {}

    def agent_a():
        r0 = a.join(0)
        results["a0"] = r0
        # simulate the agent main loop noticing the restart signal
        import time
        deadline = time.time() + 30
        while a.restart_round() <= 0:
            if time.time() > deadline:
                return
            time.sleep(0.05)
        results["a1"] = a.join(a.restart_round())

    t = threading.Thread(target=agent_a, daemon=True)
    t.start()
    # wait until round 0 is closed (world=1), then join late
    import time
    deadline = time.time() + 30
    while "a0" not in results:
        if time.time() > deadline:
            raise AssertionError("agent A never joined round 0")
        time.sleep(0.05)
    assert results["a0"].num_nodes == 1

    rb = b.join(0)  # late: round 0 already closed
    t.join(timeout=30)
    assert rb.round >= 1
    assert rb.num_nodes == 2
    assert "a1" in results
    assert results["a1"].round == rb.round
    assert {results["a1"].node_rank, rb.node_rank} == {0, 1}
    a._store = None
    b._store = None


def test_rendezvous_late_joiner_standby_replaces_failed_node():
    # Gang full (max_nodes reached): the late joiner stands by; when a
    # failure bumps the restart counter it joins the next round alone
    # (replacement-node case).
    import threading
    import time

    from torchx_amd.agent.rendezvous import C10dRendezvous, free_port

    port = free_port()
    a = C10dRendezvous(f"127.0.0.1:{port}", "unit-sb", 1, 1,
                       timeout=30.0, last_call_timeout=0.1)
    b = C10dRendezvous(f"127.0.0.1:{port}", "unit-sb", 1, 1,
                       timeout=30.0, last_call_timeout=0.1)
    ra = a.join(0)
    assert ra.num_nodes == 1

    def fail_later():
        time.sleep(0.5)
        a.signal_restart(0)  # simulates worker failure on node A

    t = threading.Thread(target=fail_later, daemon=True)
    t.start()
    rb = b.join(0)  # stands by until the restart opens round 1
    t.join(timeout=10)
    assert rb.round == 1
    assert rb.num_nodes == 1
    assert rb.node_rank == 0
    a._store = None
    b._store = None


def test_free_port_is_bindable():
    import socket

    from torchx_amd.agent.rendezvous import free_port

    p = free_port()
    with socket.socket() as s:
        s.bind(("127.0.0.1", p))
