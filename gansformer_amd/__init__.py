"""gansformer_amd — an MI355X-native GANsformer training/generation framework.

A from-scratch rebuild of the capability surface of
GiorgiaAuroraAdorni/gansformer-reproducibility-challenge (a TF1
StyleGAN2-derived GANsformer reproducibility study):
StyleGAN2 backbone (modulated conv2d, upfirdn2d, fused bias+leaky-ReLU,
minibatch-stddev) plus bipartite Simplex/Duplex attention between k latent
components and H*W image tokens, trained with logistic GAN loss, R1 and
path-length regularization, data-parallel over RCCL/xGMI.

Compute path: PyTorch-ROCm host code + hand-written CDNA4 (gfx950) HIP
kernels in `csrc/` exposed through the in-tree `gansformer_amd._C`
extension. No CUDA shims, no Triton, single dispatch path.
"""

from .config import EasyDict  # noqa: F401

__version__ = "0.1.0"
