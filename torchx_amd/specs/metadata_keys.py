"""Wire-contract metadata keys (reference parity: torchx/specs/metadata_keys.py).

Keys stored in ``AppDef.metadata``/``Role.metadata`` that cross the
launcher↔scheduler boundary and must stay stable.
"""

# the torchx context (session) name that submitted the app
CONTEXT = "torchx/context"

# launcher version that produced the request
VERSION = "torchx/version"

# experiment/run name from --name (StructuredNameArgument)
EXPERIMENT_NAME = "torchx/experiment-name"
RUN_NAME = "torchx/run-name"


def app_metadata(context: str, version: str) -> dict:
    """Standard metadata stamped on every submitted AppDef."""
    return {CONTEXT: context, VERSION: version}
