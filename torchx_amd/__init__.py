"""torchx_amd — an MI355X-native distributed job launcher and training stack.

Capabilities of meta-pytorch/torchx (job specs, pluggable schedulers, runner
session API, CLI, components, tracker), rebuilt from scratch for AMD MI355X
nodes: HIP device enumeration/pinning, RCCL process groups over xGMI, and a
bundled reference training stack whose hot ops are hand-written CDNA4 HIP
kernels (torchx_amd.ops / torchx_amd.models).
"""

__version__ = "0.1.0"
