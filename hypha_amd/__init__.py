"""hypha_amd — MI355X-native DiLoCo low-communication training framework.

A from-scratch AMD MI355X (gfx950/CDNA4) implementation of the capabilities of
hypha-space/hypha: DiLoCo training (N independent worker peers,
H local inner-AdamW steps, outer Nesterov aggregation), an auction/lease-based
control plane, data-slice serving, and a job-bridge executor API.

Compute path: PyTorch-ROCm for library GEMMs (hipBLASLt) + hand-written HIP
CDNA4 kernels for the fused hot ops (attention, RMSNorm, RoPE, SwiGLU,
cross-entropy, inner-AdamW, outer-Nesterov) + RCCL over xGMI for the outer
pseudo-gradient all-reduce and weight broadcast.
"""

__version__ = "0.1.0"
