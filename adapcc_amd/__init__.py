"""adapcc_amd — MI355X-native adaptive collective-communication framework.

A from-scratch rebuild of the capabilities of JoeyYoung/adapcc (reference
mounted read-only at /root/reference) designed for AMD Instinct MI355X
(gfx950): hand-written HIP/CDNA4 kernels pulling peer GPU memory directly
over xGMI (hipIpc), device-side chunk pipelining with system-scope flag
inboxes, RCCL/torch.distributed for bootstrap and fallbacks, and a
profile-driven strategy synthesizer that emits link-disjoint parallel star
forests for the fully connected 8-GPU mesh.
"""

from .adapcc import AdapCC
from .communicator import CommArgs, Communicator
from .primitives import Primitive

__version__ = "0.1.0"

__all__ = ["AdapCC", "CommArgs", "Communicator", "Primitive", "__version__"]
