"""Docker workspace: build a patched image on top of the role's image.

Behavior parity with the reference (torchx/workspace/docker_workspace.py:40-271):
the workspace directory is tarred (honoring ``.dockerignore``/``.torchxignore``)
into a build context; ``Dockerfile.torchx`` in the workspace is used when
present, else a default ``FROM $IMAGE; COPY . .`` Dockerfile is generated;
``role.image`` becomes the built ``sha256:...`` id. For remote schedulers,
``dryrun_push_images`` rewrites local image hashes to ``image_repo:hash`` and
``push_images`` tags + pushes them.

The ROCm twist lives in the *scheduler*, not here: containers launched from
these images get ``/dev/kfd`` + ``/dev/dri`` and ``HIP_VISIBLE_DEVICES``
pinning (see docker_scheduler.py).
"""

from __future__ import annotations

import io
import logging
import os
import tarfile
import tempfile
from typing import IO, Any, Dict, Mapping, Optional, Tuple

from torchx_amd.specs import AppDef, Role

from .api import WorkspaceMixin, walk_workspace

log = logging.getLogger(__name__)

TORCHX_DOCKERFILE = "Dockerfile.torchx"

DEFAULT_DOCKERFILE = b"""
ARG IMAGE
FROM $IMAGE

COPY . .
"""


class BuildError(Exception):
    def __init__(self, reason: str):
        super().__init__(f"docker build failed: {reason}")
        self.reason = reason


def _build_context(image: str, workspace: str) -> IO[bytes]:
    """Tar the workspace (plus a Dockerfile if none present) for docker build."""
    f = tempfile.NamedTemporaryFile(prefix="torchx-context", suffix=".tar")
    with tarfile.open(fileobj=f, mode="w") as tf:
        for abs_path, rel_path in walk_workspace(workspace):
            tf.add(abs_path, arcname=rel_path, recursive=False)
        if TORCHX_DOCKERFILE not in tf.getnames():
            info = tarfile.TarInfo(TORCHX_DOCKERFILE)
            info.size = len(DEFAULT_DOCKERFILE)
            tf.addfile(info, io.BytesIO(DEFAULT_DOCKERFILE))
    f.seek(0)
    return f


class DockerWorkspaceMixin(WorkspaceMixin):
    """Builds patched Docker images from the workspace (requires a local
    docker daemon; remote schedulers additionally need ``image_repo``)."""

    LABEL_VERSION = "torchx.amd/version"

    def __init__(self, *args: Any,
                 docker_client: Optional[Any] = None, **kwargs: Any) -> None:
        super().__init__(*args, **kwargs)
        self.__docker_client = docker_client

    @property
    def _docker_client(self) -> Any:
        if self.__docker_client is None:
            import docker

            self.__docker_client = docker.from_env()
        return self.__docker_client

    def workspace_opts(self):
        from torchx_amd.specs import runopts

        opts = runopts()
        opts.add("image_repo", type_=str,
                 help="(remote jobs) repo to push patched images to")
        opts.add("quiet", type_=bool, default=False,
                 help="suppress docker build output")
        return opts

    def build_workspace_and_update_role(self, role: Role, workspace: str,
                                        cfg: Mapping[str, Any]) -> str:
        context = _build_context(role.image, workspace)
        try:
            try:
                self._docker_client.images.pull(role.image)
            except Exception as e:  # noqa: BLE001 — local-only images are fine
                log.warning("failed to pull `%s`, using local: %s",
                            role.image, e)
            log.info("building workspace docker image...")
            events = self._docker_client.api.build(
                fileobj=context,
                custom_context=True,
                dockerfile=TORCHX_DOCKERFILE,
                buildargs={"IMAGE": role.image, "WORKSPACE": workspace},
                pull=False,
                rm=True,
                decode=True,
                labels={self.LABEL_VERSION: "0.1.0"},
            )
            image_id = None
            for event in events:
                if (msg := event.get("stream")) and not cfg.get("quiet"):
                    if msg.strip():
                        log.info(msg.strip())
                if aux := event.get("aux"):
                    image_id = aux["ID"]
                if error := event.get("error"):
                    raise BuildError(error)
            if not image_id:
                raise BuildError("no image id in build output")
            role.image = image_id
            return image_id
        finally:
            context.close()

    def dryrun_push_images(self, app: AppDef,
                           cfg: Mapping[str, Any]) -> Dict[str, Tuple[str, str]]:
        """Rewrite local ``sha256:`` images to ``image_repo:hash``; returns
        the ``{local: (repo, tag)}`` map that :meth:`push_images` consumes."""
        prefix = "sha256:"
        image_repo = cfg.get("image_repo")
        to_push: Dict[str, Tuple[str, str]] = {}
        for role in app.roles:
            if role.image.startswith(prefix):
                if not image_repo:
                    raise KeyError(
                        f"set the `image_repo` config to push local image "
                        f"{role.image}"
                    )
                tag = role.image[len(prefix):]
                to_push[role.image] = (str(image_repo), tag)
                role.image = f"{image_repo}:{tag}"
        return to_push

    def push_images(self, images: Optional[Dict[str, Tuple[str, str]]]) -> None:
        if not images:
            return
        client = self._docker_client
        for local, (repo, tag) in images.items():
            log.info("pushing %s:%s", repo, tag)
            img = client.images.get(local)
            img.tag(repo, tag=tag)
            client.images.push(repo, tag=tag)
