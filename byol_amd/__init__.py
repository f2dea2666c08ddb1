"""byol_amd — MI355X-native BYOL self-supervised pretraining engine.

A from-scratch framework with the capabilities of jramapuram/BYOL
(reference at /root/reference), built MI355X-first: PyTorch-ROCm +
hand-written CDNA4 (gfx950) HIP kernels for the hot ops + RCCL over xGMI
for data parallelism.
"""

__version__ = "0.2.0"  # round 2: measured conv/BN dispatch, bf16 BN, aug v2, HYBRID find
