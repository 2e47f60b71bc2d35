"""Process monitor app: runs a target command and reports exit/resource
status (parity: torchx/apps/utils/process_monitor.py).  Used by schedulers
that need a supervising shim around a non-cooperative binary."""

from __future__ import annotations

import argparse
import subprocess
import sys
import time


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description="supervise a child command")
    p.add_argument("--timeout", type=float, default=None,
                   help="kill the child after this many seconds")
    p.add_argument("--poll", type=float, default=1.0)
    p.add_argument("cmd", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)
    cmd = args.cmd
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        print("no command given", file=sys.stderr)
        return 2
    start = time.time()
    proc = subprocess.Popen(cmd)
    while True:
        rc = proc.poll()
        if rc is not None:
            print(f"child exited rc={rc} after {time.time()-start:.1f}s",
                  flush=True)
            return rc
        if args.timeout and time.time() - start > args.timeout:
            proc.terminate()
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()
            print("child timed out", file=sys.stderr)
            return 124
        time.sleep(args.poll)


if __name__ == "__main__":
    sys.exit(main())
