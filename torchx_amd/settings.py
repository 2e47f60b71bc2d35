"""Centralized environment-variable name constants (parity:
torchx/settings.py:22-34). The wire contract these names form is what
keeps existing torchx-style workflows running unchanged;
modules reference them from here instead of re-typing strings."""

# tracker / lineage (consumed by tracker.api.AppRun.run_from_env)
ENV_TORCHX_TRACKERS = "TORCHX_TRACKERS"
ENV_TORCHX_PARENT_RUN_ID = "TORCHX_PARENT_RUN_ID"
ENV_TORCHX_JOB_ID = "TORCHX_JOB_ID"

# config discovery (runner.config)
ENV_TORCHXCONFIG = "TORCHXCONFIG"

# session (runner telemetry correlation)
ENV_TORCHX_INTERNAL_SESSION_ID = "TORCHX_INTERNAL_SESSION_ID"

# scheduler-set replica env
ENV_TORCHX_IMAGE = "TORCHX_IMAGE"
ENV_TORCHX_RANK0_HOST = "TORCHX_RANK0_HOST"
ENV_TORCHX_CONTEXT_NAME = "TORCHX_CONTEXT_NAME"

# torchelastic integration (agent <-> scheduler reply files)
ENV_TORCHELASTIC_ERROR_FILE = "TORCHELASTIC_ERROR_FILE"
ENV_PET_LOG_DIR = "PET_LOG_DIR"

# MI355X device pinning (the CUDA_VISIBLE_DEVICES analogs)
ENV_HIP_VISIBLE_DEVICES = "HIP_VISIBLE_DEVICES"
ENV_ROCR_VISIBLE_DEVICES = "ROCR_VISIBLE_DEVICES"
