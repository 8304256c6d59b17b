"""Op layer: Python wrappers over the gfx950 HIP extension with eager
(pure-PyTorch) reference fallbacks for CPU.

Dispatch policy (single path, no multi-backend machinery):
  * tensor on HIP device  -> gansformer_amd._C kernels. If the extension
    is missing on a GPU machine the op RAISES — no silent eager fallback.
  * tensor on CPU         -> eager reference implementation (also the
    golden numerics reference for kernel tests).
"""

from .fused_act import fused_bias_act, bias_act, mod_bias_act  # noqa: F401
from .upfirdn2d import (  # noqa: F401
    upfirdn2d,
    setup_filter,
    upsample2d,
    downsample2d,
    filter2d,
)
from .conv2d_grad import conv2d_gradfix  # noqa: F401
from .modulated_conv import modulated_conv2d  # noqa: F401
from .mbstd import minibatch_stddev  # noqa: F401
from .modnorm import modnorm  # noqa: F401
from .linear import linear_nobias, linear_transposed  # noqa: F401
from .bipartite import bipartite_attention  # noqa: F401
