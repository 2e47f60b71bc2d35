"""Directory workspaces (parity: torchx/workspace/dir_workspace.py):
copy the workspace into a job dir (slurm) or tmp dir and point role.image
at it."""

from __future__ import annotations

import os
import shutil
import tempfile
from typing import Any, Mapping

from torchx_amd.specs import Role

from .api import WorkspaceMixin, walk_workspace


def _copy_workspace(workspace: str, dst: str) -> None:
    for abs_path, rel in walk_workspace(workspace):
        target = os.path.join(dst, rel)
        os.makedirs(os.path.dirname(target), exist_ok=True)
        shutil.copy2(abs_path, target)


class TmpDirWorkspaceMixin(WorkspaceMixin):
    def build_workspace_and_update_role(self, role: Role, workspace: str,
                                        cfg: Mapping[str, Any]) -> str:
        job_dir = tempfile.mkdtemp(prefix="torchx_amd_workspace_")
        _copy_workspace(workspace, job_dir)
        role.image = job_dir
        return job_dir


class DirWorkspaceMixin(WorkspaceMixin):
    def build_workspace_and_update_role(self, role: Role, workspace: str,
                                        cfg: Mapping[str, Any]) -> str:
        job_dir = cfg.get("job_dir")
        if not job_dir:
            return role.image
        os.makedirs(job_dir, exist_ok=True)
        _copy_workspace(workspace, job_dir)
        role.image = job_dir
        return job_dir
