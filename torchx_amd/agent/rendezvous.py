"""c10d TCPStore-based rendezvous for the torchx_amd elastic agent.

Replaces the torchelastic rendezvous the reference delegates to
(torchx/components/dist.py:262 invokes `torchrun --rdzv_backend c10d`).
Protocol (per restart round r, keys under ``rdzv/{run_id}/{r}/``):

  1. every agent joins with store.add(prefix+"count") -> its node seq
  2. the first joiner (seq 0) is the round leader: it waits until
     count >= min_nodes, then grace-waits for stragglers up to max_nodes,
     publishes {world: n, master_addr, master_port} and closes the round
  3. everyone else waits for the "closed" key
  4. a worker failure anywhere bumps ``rdzv/{run_id}/restart``; agents poll
     it between worker waits and re-join round r+1 after tearing down.

The TCPStore server is hosted by the agent whose host matches the
rendezvous endpoint (every agent falls back to client if the bind fails).
"""

from __future__ import annotations

import json
import logging
import socket
import time
from dataclasses import dataclass
from datetime import timedelta
from typing import Optional, Tuple

log = logging.getLogger(__name__)


def _local_hostnames() -> set:
    names = {"localhost", "127.0.0.1", "0.0.0.0", socket.gethostname()}
    try:
        names.add(socket.getfqdn())
    except OSError:
        pass
    try:
        for info in socket.getaddrinfo(socket.gethostname(), None):
            names.add(info[4][0])
    except OSError:
        pass
    return names


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("", 0))
        return s.getsockname()[1]


@dataclass
class RendezvousResult:
    round: int
    node_rank: int
    num_nodes: int
    master_addr: str
    master_port: int


class C10dRendezvous:
    def __init__(self, endpoint: str, run_id: str, min_nodes: int,
                 max_nodes: int, timeout: float = 600.0,
                 last_call_timeout: float = 5.0) -> None:
        host, _, port = endpoint.partition(":")
        self.host = host or "localhost"
        self.port = int(port or 29500)
        self.run_id = run_id
        self.min_nodes = min_nodes
        self.max_nodes = max_nodes
        self.timeout = timeout
        self.last_call_timeout = last_call_timeout
        self._store = None
        self._is_host = self.host in _local_hostnames()

    def store(self):
        if self._store is None:
            from torch.distributed import TCPStore

            deadline = time.time() + self.timeout
            last_err: Optional[Exception] = None
            while time.time() < deadline:
                # the host agent serves the store; everyone else connects.
                for is_master in ((True, False) if self._is_host else (False,)):
                    try:
                        self._store = TCPStore(
                            self.host, self.port, is_master=is_master,
                            timeout=timedelta(seconds=min(30.0, self.timeout)),
                            wait_for_workers=False,
                        )
                        return self._store
                    except Exception as e:  # noqa: BLE001 (bind race / conn refused)
                        last_err = e
                time.sleep(1.0)
            raise TimeoutError(
                f"could not reach rendezvous store {self.host}:{self.port}: {last_err}"
            )
        return self._store

    # -- restart signalling -------------------------------------------------
    def restart_round(self) -> int:
        return int(self.store().add(f"rdzv/{self.run_id}/restart", 0))

    def signal_restart(self, current_round: int) -> int:
        """Bump the restart counter to current_round+1 (idempotent-ish)."""
        store = self.store()
        key = f"rdzv/{self.run_id}/restart"
        val = int(store.add(key, 0))
        if val <= current_round:
            val = int(store.add(key, current_round + 1 - val))
        return val

    # -- join ---------------------------------------------------------------
    def join(self, round_: int) -> RendezvousResult:
        """Join round ``round_`` (or a later one).

        A node that arrives after a round has closed does NOT error out
        (torchelastic semantics — the reference delegates this to
        torchelastic via torchx/components/dist.py:262): if the closed
        round has room below max_nodes the late joiner signals a
        re-rendezvous (elastic scale-up) and everyone re-forms at the next
        round; if the gang is already full it stands by until the next
        restart round opens. The returned ``RendezvousResult.round`` is
        therefore authoritative and may be greater than ``round_``.
        """
        deadline = time.time() + self.timeout
        while True:
            result = self._try_join(round_)
            if result is not None:
                return result
            # round closed before we joined
            world = self._round_world(round_)
            if world is not None and world < self.max_nodes:
                # scale-up: ask the running gang to re-rendezvous with us
                log.info(
                    "late join for round %d (world %d < max %d); "
                    "signalling scale-up re-rendezvous", round_, world,
                    self.max_nodes,
                )
                self.signal_restart(round_)
            round_ = self._wait_for_round_after(round_, deadline)

    def _round_world(self, round_: int) -> Optional[int]:
        store = self.store()
        prefix = f"rdzv/{self.run_id}/{round_}/"
        try:
            if store.check([prefix + "closed"]):
                return int(json.loads(store.get(prefix + "closed"))["world"])
        except Exception:  # noqa: BLE001
            pass
        return None

    def _wait_for_round_after(self, round_: int, deadline: float) -> int:
        """Standby: poll the restart counter until a round > round_ opens."""
        while True:
            cur = self.restart_round()
            if cur > round_:
                return cur
            if time.time() > deadline:
                raise TimeoutError(
                    f"timed out standing by for a rendezvous round after "
                    f"{round_}"
                )
            time.sleep(0.5)

    def _try_join(self, round_: int) -> Optional[RendezvousResult]:
        store = self.store()
        prefix = f"rdzv/{self.run_id}/{round_}/"
        seq = int(store.add(prefix + "count", 1)) - 1
        if seq == 0:
            # leader
            deadline = time.time() + self.timeout
            while int(store.add(prefix + "count", 0)) < self.min_nodes:
                if time.time() > deadline:
                    raise TimeoutError(
                        f"rendezvous round {round_}: only "
                        f"{int(store.add(prefix + 'count', 0))} of "
                        f"{self.min_nodes} nodes joined"
                    )
                time.sleep(0.1)
            # grace period for stragglers up to max_nodes
            grace_end = time.time() + self.last_call_timeout
            n = int(store.add(prefix + "count", 0))
            while n < self.max_nodes and time.time() < grace_end:
                time.sleep(0.1)
                n = int(store.add(prefix + "count", 0))
            n = min(n, self.max_nodes)
            # global rank 0 lives on the leader node, so MASTER_ADDR must
            # resolve to it.  Prefer the (known-resolvable) endpoint host
            # when the leader IS the endpoint host; else our hostname.
            if self._is_host and self.host != "0.0.0.0":
                addr = self.host
            else:
                addr = socket.gethostname()
            info = {
                "world": n,
                "master_addr": addr,
                "master_port": free_port(),
            }
            store.set(prefix + "closed", json.dumps(info))
        blob = store.get(prefix + "closed")
        info = json.loads(blob)
        if seq >= info["world"]:
            # joined after the leader closed the round — caller retries
            return None
        return RendezvousResult(
            round=round_,
            node_rank=seq,
            num_nodes=info["world"],
            master_addr=info["master_addr"],
            master_port=info["master_port"],
        )
