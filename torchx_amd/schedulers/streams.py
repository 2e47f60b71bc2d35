"""Non-blocking fd tee for combined replica logs (parity:
torchx/schedulers/streams.py:16)."""

from __future__ import annotations

import threading
from typing import BinaryIO, List


class Tee:
    """Copies everything read from ``src_fds`` into ``dst`` (binary file),
    one reader thread per source."""

    def __init__(self, dst: BinaryIO, *src_paths: str) -> None:
        self._dst = dst
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._positions = {p: 0 for p in src_paths}
        for p in src_paths:
            t = threading.Thread(target=self._pump, args=(p,), daemon=True)
            t.start()
            self._threads.append(t)

    def _pump(self, path: str) -> None:
        pos = 0
        while True:
            try:
                with open(path, "rb") as f:
                    f.seek(pos)
                    data = f.read(65536)
            except OSError:
                data = b""
            if data:
                pos += len(data)
                with self._lock:
                    self._dst.write(data)
                    self._dst.flush()
            elif self._stop.is_set():
                break
            else:
                self._stop.wait(0.1)

    def close(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        with self._lock:
            try:
                self._dst.flush()
                self._dst.close()
            except ValueError:
                pass
