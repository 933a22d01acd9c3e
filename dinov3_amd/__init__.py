"""dinov3_amd: MI355X-native DINOv3 self-supervised pretraining framework.

PyTorch-ROCm host + hand-written CDNA4 (gfx950) HIP kernels + RCCL over xGMI.
Capability parity target: Dhia-naouali/dinov3-jax.
"""

__version__ = "0.1.0"
