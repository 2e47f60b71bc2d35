"""Simple KV result tracker for app results, e.g. HPO objective values
(parity: torchx/runtime/tracking/api.py: ResultTracker :19,
FsspecResultTracker :125)."""

from __future__ import annotations

import json
from typing import Any, Dict, Union

import fsspec


class ResultTracker:
    def __setitem__(self, key: Union[int, str], value: Dict[str, Any]) -> None:
        raise NotImplementedError

    def __getitem__(self, key: Union[int, str]) -> Dict[str, Any]:
        raise NotImplementedError


class FsspecResultTracker(ResultTracker):
    """Stores each entry as ``<base>/<key>/result.json`` over fsspec."""

    def __init__(self, tracker_base: str) -> None:
        self._base = str(tracker_base).rstrip("/")

    def _path(self, key: Union[int, str]) -> str:
        return f"{self._base}/{key}/result.json"

    def __setitem__(self, key: Union[int, str], value: Dict[str, Any]) -> None:
        path = self._path(key)
        fs, _, _ = fsspec.get_fs_token_paths(path)
        fs.makedirs(f"{self._base}/{key}", exist_ok=True)
        with fsspec.open(path, "w") as f:
            json.dump(value, f)

    def __getitem__(self, key: Union[int, str]) -> Dict[str, Any]:
        with fsspec.open(self._path(key), "r") as f:
            return json.load(f)
