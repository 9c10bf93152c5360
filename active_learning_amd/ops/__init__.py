"""MI355X-native op layer.

``functional`` exposes autograd-integrated ops. On CUDA (= ROCm/HIP) tensors
they dispatch to the in-tree HIP/CDNA4 extension (``active_learning_amd._C``,
gfx950 kernels under ops/hip/); on CPU tensors they fall back to plain PyTorch
implementations that define the reference semantics used by the unit tests.

There is deliberately NO torch fallback on GPU: if the extension is missing on
a GPU machine the ops raise, so a silently-eager path can never masquerade as
the native one.
"""

from . import functional  # noqa: F401
from .extension import extension_available, load_extension  # noqa: F401
