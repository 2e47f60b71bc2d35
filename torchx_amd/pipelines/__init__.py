"""Pipeline adapters namespace (reference parity: torchx/pipelines/__init__.py:7-14).

Adapters transform components (AppDefs) into stages of external pipeline
engines (e.g. KFP). The namespace is intentionally empty in-core: providers
ship adapters as plugins under ``torchx_plugins``.
"""
