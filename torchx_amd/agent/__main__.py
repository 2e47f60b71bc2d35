"""torchx_amd elastic agent — the native torchrun replacement (the
reference invokes `torchrun` as a subprocess, torchx/components/dist.py:262,
delegating all of this to torchelastic).

Launched by the dist.ddp component (one agent per node), it:
  * joins a c10d TCPStore rendezvous (rendezvous.py),
  * spawns one worker process per local GPU (or --nproc-per-node),
  * wires the torch.distributed env contract (RANK/LOCAL_RANK/WORLD_SIZE/
    MASTER_ADDR/MASTER_PORT/...) so workers bring up RCCL over xGMI,
  * supervises workers: on failure, signals every agent through the store,
    tears down, re-rendezvouses and restarts (up to --max-restarts),
  * propagates the first failure's torchelastic error file to the
    scheduler's TORCHELASTIC_ERROR_FILE reply file.

Usage:
  python -m torchx_amd.agent --nnodes 1:1 --nproc-per-node 8 \
      --rdzv-endpoint localhost:29500 --rdzv-id app_1 --max-restarts 0 \
      [--tee] [--no-python] script.py [script args...]
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import shutil
import signal
import subprocess
import sys
import threading
import time
import traceback
from typing import Dict, List, Optional

from .rendezvous import C10dRendezvous, RendezvousResult

log = logging.getLogger("torchx_amd.agent")

POLL_INTERVAL = 1.0


def parse_nnodes(spec: str):
    if ":" in spec:
        lo, hi = spec.split(":")
        return int(lo), int(hi)
    n = int(spec)
    return n, n


class Worker:
    def __init__(self, proc: subprocess.Popen, local_rank: int,
                 global_rank: int, error_file: str,
                 pumps: List[threading.Thread]) -> None:
        self.proc = proc
        self.local_rank = local_rank
        self.global_rank = global_rank
        self.error_file = error_file
        self.pumps = pumps


def _pump(src, dst, prefix: str,
          file_path: Optional[str] = None) -> threading.Thread:
    """Prefix-tee a worker stream to the agent's stream AND (torchrun
    --tee parity) a per-rank file under the log dir."""

    def run() -> None:
        f = None
        if file_path:
            try:
                os.makedirs(os.path.dirname(file_path), exist_ok=True)
                f = open(file_path, "ab")
            except OSError:
                f = None
        for line in iter(src.readline, b""):
            try:
                dst.buffer.write(prefix.encode() + line)
                dst.flush()
            except ValueError:
                break
            if f is not None:
                try:
                    f.write(line)
                    f.flush()
                except OSError:
                    f = None
        src.close()
        if f is not None:
            f.close()

    t = threading.Thread(target=run, daemon=True)
    t.start()
    return t


def _default_nproc() -> int:
    from torchx_amd.schedulers.devices import hip_device_count

    n = hip_device_count()
    return n if n > 0 else 1


def worker_numa_prefix(local_rank: int,
                       env: Optional[Dict[str, str]] = None) -> List[str]:
    """numactl argv prefix binding worker ``local_rank`` to the NUMA node
    of ITS GPU (SURVEY §7 step 2 at worker granularity — the replica-level
    binding no-ops when one agent owns GPUs on several nodes).  Resolves
    the worker's GLOBAL device index through HIP_VISIBLE_DEVICES."""
    from torchx_amd.schedulers.devices import numa_bind_args

    env = env if env is not None else os.environ
    visible = env.get("HIP_VISIBLE_DEVICES") or env.get(
        "ROCR_VISIBLE_DEVICES")
    if visible:
        devs = [d.strip() for d in visible.split(",") if d.strip()]
        if local_rank >= len(devs):
            return []
        dev = devs[local_rank]
    else:
        dev = str(local_rank)
    return numa_bind_args(dev)


def worker_cpu_affinity(local_rank: int,
                        env: Optional[Dict[str, str]] = None) -> str:
    """cpulist of the worker GPU's NUMA node ("" if unknown): the
    numactl-free affinity path, exported as TORCHX_AMD_CPU_AFFINITY and
    applied by torchx_amd.distributed.init_pg via sched_setaffinity
    (numactl is not installed on the target fleet)."""
    from torchx_amd.schedulers.devices import numa_cpulist, numa_node_of

    env = env if env is not None else os.environ
    visible = env.get("HIP_VISIBLE_DEVICES") or env.get(
        "ROCR_VISIBLE_DEVICES")
    if visible:
        devs = [d.strip() for d in visible.split(",") if d.strip()]
        if local_rank >= len(devs):
            return ""
        dev = devs[local_rank]
    else:
        dev = str(local_rank)
    return numa_cpulist(numa_node_of(dev))


def spawn_workers(args, rdzv: RendezvousResult, restart_count: int,
                  log_dir: str) -> List[Worker]:
    nproc = args.nproc_per_node
    world_size = rdzv.num_nodes * nproc
    workers: List[Worker] = []
    cmd_base: List[str]
    if args.no_python:
        cmd_base = [args.script]
    else:
        cmd_base = [sys.executable, args.script]
    for lr in range(nproc):
        prof_prefix: List[str] = []
        if args.rocprof:
            prof_dir = os.path.join(log_dir, f"rocprof_rank{lr}")
            prof_prefix = ["rocprofv3", "--kernel-trace", "--stats",
                           "-d", prof_dir, "-o", "prof", "--"]
        grank = rdzv.node_rank * nproc + lr
        error_file = os.path.join(log_dir, f"worker_{grank}_error.json")
        if os.path.exists(error_file):
            os.unlink(error_file)
        env = {
            **os.environ,
            "RANK": str(grank),
            "LOCAL_RANK": str(lr),
            "WORLD_SIZE": str(world_size),
            "LOCAL_WORLD_SIZE": str(nproc),
            "GROUP_RANK": str(rdzv.node_rank),
            "ROLE_RANK": str(grank),
            "ROLE_WORLD_SIZE": str(world_size),
            "MASTER_ADDR": rdzv.master_addr,
            "MASTER_PORT": str(rdzv.master_port),
            "TORCHELASTIC_RESTART_COUNT": str(restart_count),
            "TORCHELASTIC_MAX_RESTARTS": str(args.max_restarts),
            "TORCHELASTIC_RUN_ID": args.rdzv_id,
            "TORCHELASTIC_ERROR_FILE": error_file,
            "OMP_NUM_THREADS": os.environ.get("OMP_NUM_THREADS", "1"),
        }
        if args.numa_affinity:
            aff = worker_cpu_affinity(lr)
            if aff:
                env["TORCHX_AMD_CPU_AFFINITY"] = aff
        stdout = subprocess.PIPE if args.tee else None
        stderr = subprocess.PIPE if args.tee else None
        numa_prefix = worker_numa_prefix(lr) if args.numa_affinity else []
        proc = subprocess.Popen(
            numa_prefix + prof_prefix + cmd_base + args.script_args,
            env=env,
            stdout=stdout,
            stderr=stderr,
            start_new_session=True,
        )
        pumps = []
        if args.tee:
            # per-rank files under the log dir (torchrun --tee/PET_LOG_DIR
            # layout: <log_dir>/<restart>/<local_rank>/std{out,err}.log)
            rank_dir = os.path.join(log_dir, str(restart_count), str(lr))
            pumps.append(_pump(proc.stdout, sys.stdout, f"[rank{grank}]: ",
                               os.path.join(rank_dir, "stdout.log")))
            pumps.append(_pump(proc.stderr, sys.stderr, f"[rank{grank}]: ",
                               os.path.join(rank_dir, "stderr.log")))
        workers.append(Worker(proc, lr, grank, error_file, pumps))
    return workers


def kill_workers(workers: List[Worker]) -> None:
    for w in workers:
        if w.proc.poll() is None:
            try:
                os.killpg(w.proc.pid, signal.SIGTERM)
            except ProcessLookupError:
                pass
    deadline = time.time() + 10
    for w in workers:
        while w.proc.poll() is None and time.time() < deadline:
            time.sleep(0.1)
        if w.proc.poll() is None:
            try:
                os.killpg(w.proc.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
            w.proc.wait()


def propagate_error(workers: List[Worker]) -> Optional[dict]:
    """Earliest worker error file wins; copied to the scheduler reply file."""
    best = None
    best_mtime = float("inf")
    for w in workers:
        if os.path.isfile(w.error_file):
            mt = os.path.getmtime(w.error_file)
            if mt < best_mtime:
                best_mtime = mt
                best = w.error_file
    reply = os.environ.get("TORCHELASTIC_ERROR_FILE")
    if best and reply and best != reply:
        os.makedirs(os.path.dirname(reply) or ".", exist_ok=True)
        shutil.copyfile(best, reply)
    if best:
        try:
            with open(best) as f:
                return json.load(f)
        except (OSError, json.JSONDecodeError):
            return None
    return None


def write_agent_error(msg: str) -> None:
    reply = os.environ.get("TORCHELASTIC_ERROR_FILE")
    if not reply:
        return
    try:
        os.makedirs(os.path.dirname(reply) or ".", exist_ok=True)
        with open(reply, "w") as f:
            json.dump(
                {"message": {"message": msg,
                             "extraInfo": {"timestamp": str(int(time.time()))}}},
                f,
            )
    except OSError:
        pass


def main(argv: Optional[List[str]] = None) -> int:
    p = argparse.ArgumentParser(prog="torchx_amd.agent")
    p.add_argument("--nnodes", type=str, default="1:1")
    p.add_argument("--nproc-per-node", type=str, default="auto")
    p.add_argument("--rdzv-endpoint", type=str, default="localhost:29500")
    p.add_argument("--rdzv-id", type=str, default="default")
    p.add_argument("--max-restarts", type=int, default=0)
    p.add_argument("--tee", action="store_true", default=True)
    p.add_argument("--no-tee", dest="tee", action="store_false")
    p.add_argument("--no-python", action="store_true")
    p.add_argument("--numa-affinity", action="store_true", default=True)
    p.add_argument("--no-numa-affinity", dest="numa_affinity",
                   action="store_false")
    p.add_argument("--rocprof", action="store_true",
                   help="wrap each worker in rocprofv3 --kernel-trace "
                        "--stats (output under the log dir)")
    p.add_argument("--log-dir", type=str, default=None)
    p.add_argument("script", type=str)
    p.add_argument("script_args", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)

    logging.basicConfig(
        level=os.environ.get("LOGLEVEL", "INFO"),
        format="[torchx_amd.agent] %(levelname)s: %(message)s",
    )

    if args.nproc_per_node in ("auto", "gpu"):
        args.nproc_per_node = _default_nproc()
    else:
        args.nproc_per_node = int(args.nproc_per_node)

    min_nodes, max_nodes = parse_nnodes(args.nnodes)
    log_dir = args.log_dir or os.environ.get("PET_LOG_DIR") or "/tmp"
    os.makedirs(log_dir, exist_ok=True)

    rdzv = C10dRendezvous(args.rdzv_endpoint, args.rdzv_id, min_nodes,
                          max_nodes)

    restart_count = 0
    workers: List[Worker] = []

    def handle_term(signum, frame):
        kill_workers(workers)
        sys.exit(128 + signum)

    signal.signal(signal.SIGTERM, handle_term)
    signal.signal(signal.SIGINT, handle_term)

    try:
        round_ = rdzv.restart_round()
        while True:
            log.info(
                "rendezvous round %d (restart %d/%d), nproc_per_node=%d",
                round_, restart_count, args.max_restarts, args.nproc_per_node,
            )
            result = rdzv.join(round_)
            # a late joiner may have stood by / scale-up-signalled into a
            # later round — result.round is authoritative
            round_ = result.round
            log.info(
                "joined as node %d/%d; master %s:%d",
                result.node_rank, result.num_nodes, result.master_addr,
                result.master_port,
            )
            workers = spawn_workers(args, result, restart_count, log_dir)

            failed: Optional[Worker] = None
            while True:
                time.sleep(POLL_INTERVAL)
                states = [w.proc.poll() for w in workers]
                if any(rc is not None and rc != 0 for rc in states):
                    failed = next(
                        w for w, rc in zip(workers, states)
                        if rc is not None and rc != 0
                    )
                    break
                if all(rc == 0 for rc in states):
                    log.info("all %d local workers succeeded", len(workers))
                    return 0
                # another node may have failed: check the restart counter
                cur = rdzv.restart_round()
                if cur > round_:
                    log.warning("restart signalled by another node (round %d)",
                                cur)
                    round_ = cur
                    failed = None
                    break

            kill_workers(workers)
            if failed is not None:
                log.error(
                    "worker rank %d exited rc=%d", failed.global_rank,
                    failed.proc.returncode,
                )
                round_ = rdzv.signal_restart(round_)
            restart_count += 1
            if restart_count > args.max_restarts:
                propagate_error(workers)
                log.error("exceeded max restarts (%d); failing",
                          args.max_restarts)
                return 1
    except Exception as e:  # noqa: BLE001
        kill_workers(workers)
        write_agent_error(f"agent error: {e}\n{traceback.format_exc()}")
        raise
    finally:
        kill_workers(workers)


if __name__ == "__main__":
    sys.exit(main())
