"""fsspec tracker backend (parity: torchx/tracker/backend/fsspec.py:67-280).

Layout: ``<root>/<b32(run_id)>/{artifacts,metadata,sources,descendants}/``
with one JSON file per entry; run ids are base32-encoded so arbitrary app
handles are filesystem-safe."""

from __future__ import annotations

import base64
import json
import posixpath
from typing import Iterable, Mapping, Optional

import fsspec

from .api import TrackerBase


def _enc(run_id: str) -> str:
    return base64.b32encode(run_id.encode()).decode().rstrip("=")


def _dec(enc: str) -> str:
    pad = "=" * (-len(enc) % 8)
    return base64.b32decode(enc + pad).decode()


class FsspecTracker(TrackerBase):
    def __init__(self, root: str) -> None:
        self._root = root.rstrip("/")
        self._fs, _, _ = fsspec.get_fs_token_paths(self._root or "/")

    def _dir(self, run_id: str, kind: str) -> str:
        return posixpath.join(self._root, _enc(run_id), kind)

    def _write(self, path: str, data: dict) -> None:
        self._fs.makedirs(posixpath.dirname(path), exist_ok=True)
        with fsspec.open(path, "w") as f:
            json.dump(data, f)

    # -- artifacts ----------------------------------------------------------
    def add_artifact(self, run_id: str, name: str, path: str,
                     metadata: Optional[Mapping[str, object]] = None) -> None:
        self._write(
            posixpath.join(self._dir(run_id, "artifacts"), f"{name}.json"),
            {"name": name, "path": path, "metadata": dict(metadata or {})},
        )

    def artifacts(self, run_id: str) -> Mapping[str, str]:
        d = self._dir(run_id, "artifacts")
        out = {}
        if self._fs.exists(d):
            for p in self._fs.ls(d):
                with fsspec.open(p, "r") as f:
                    e = json.load(f)
                out[e["name"]] = e["path"]
        return out

    # -- metadata -----------------------------------------------------------
    def add_metadata(self, run_id: str, **kwargs: object) -> None:
        path = posixpath.join(self._dir(run_id, "metadata"), "metadata.json")
        existing = self.metadata(run_id)
        merged = {**existing, **kwargs}
        self._write(path, merged)

    def metadata(self, run_id: str) -> Mapping[str, object]:
        path = posixpath.join(self._dir(run_id, "metadata"), "metadata.json")
        if self._fs.exists(path):
            with fsspec.open(path, "r") as f:
                return json.load(f)
        return {}

    # -- lineage ------------------------------------------------------------
    def add_source(self, run_id: str, source_id: str,
                   artifact_name: Optional[str] = None) -> None:
        name = _enc(source_id) + (f"_{artifact_name}" if artifact_name else "")
        self._write(
            posixpath.join(self._dir(run_id, "sources"), f"{name}.json"),
            {"source": source_id, "artifact": artifact_name},
        )
        # reverse edge for descendant queries
        self._write(
            posixpath.join(self._dir(source_id, "descendants"),
                           f"{_enc(run_id)}.json"),
            {"descendant": run_id},
        )

    def sources(self, run_id: str,
                artifact_name: Optional[str] = None) -> Iterable[str]:
        d = self._dir(run_id, "sources")
        out = []
        if self._fs.exists(d):
            for p in self._fs.ls(d):
                with fsspec.open(p, "r") as f:
                    e = json.load(f)
                if artifact_name is None or e.get("artifact") == artifact_name:
                    out.append(e["source"])
        return out

    def lineage(self, run_id: str) -> Iterable[str]:
        """Direct descendants of ``run_id`` (runs that declared it a
        source) — read from the reverse edges add_source writes."""
        d = self._dir(run_id, "descendants")
        out = []
        if self._fs.exists(d):
            for p in self._fs.ls(d):
                with fsspec.open(p, "r") as f:
                    out.append(json.load(f)["descendant"])
        return out

    def run_ids(self, **kwargs: str) -> Iterable[str]:
        if not self._fs.exists(self._root):
            return []
        out = []
        for p in self._fs.ls(self._root):
            name = posixpath.basename(p.rstrip("/"))
            try:
                out.append(_dec(name))
            except Exception:  # noqa: BLE001
                continue
        return out


def create(config: Optional[str]) -> FsspecTracker:
    """Factory (entry-point contract): config is the root path or a file
    containing it."""
    root = config or "/tmp/torchx-tracker"
    import os

    if os.path.isfile(root):
        with open(root) as f:
            root = f.read().strip()
    return FsspecTracker(root)
