"""Threaded multi-replica log merge with colored role/replica prefixes.

Behavior parity with the reference (torchx/util/log_tee_helpers.py:58-171,
torchx/cli/cmd_log.py:98-177): one thread per (role, replica) pulls that
replica's log iterator and prints each line prefixed with a colored
``role/replica`` tag, interleaving output as it arrives.
"""

from __future__ import annotations

import sys
import threading
from typing import Callable, IO, Iterable, List, Optional, Tuple

_COLORS = [32, 33, 34, 35, 36, 92, 93, 94, 95, 96]  # ANSI fg codes


def _prefix(role: str, replica: int, idx: int, colored: bool) -> str:
    tag = f"{role}/{replica}"
    if colored:
        return f"\033[{_COLORS[idx % len(_COLORS)]}m{tag}\033[0m "
    return f"{tag} "


def print_log_lines(
    targets: List[Tuple[str, int]],
    line_iterator: Callable[[str, int], Iterable[str]],
    stream: Optional[IO[str]] = None,
    colored: Optional[bool] = None,
) -> None:
    """Merge the log streams of ``targets`` (role, replica) onto ``stream``.

    ``line_iterator(role, replica)`` yields that replica's lines; one
    daemon thread per target pulls and prints with a stable colored
    prefix. Exceptions in a puller are re-raised in the caller.
    """
    out = stream if stream is not None else sys.stdout
    use_color = colored if colored is not None else out.isatty()
    lock = threading.Lock()
    errors: List[BaseException] = []

    def pull(role: str, replica: int, idx: int) -> None:
        pfx = _prefix(role, replica, idx, use_color)
        try:
            for line in line_iterator(role, replica):
                with lock:
                    out.write(pfx + line.rstrip("\n") + "\n")
                    out.flush()
        except BaseException as e:  # noqa: BLE001 — surfaced to caller
            errors.append(e)

    threads = [
        threading.Thread(target=pull, args=(r, k, i), daemon=True)
        for i, (r, k) in enumerate(targets)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    if errors:
        raise errors[0]
