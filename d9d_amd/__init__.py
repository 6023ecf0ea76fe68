"""d9d_amd — an MI355X-native (CDNA4/gfx950) composable distributed-training framework.

Brand-new implementation of the capabilities of the d9d reference
(`/root/reference`, d9d-project/d9d v0.16.0): composable DP/FSDP/HSDP/TP/SP/PP/EP/CP
parallelism over torch DeviceMesh/DTensor, an in-house pipeline engine, MoE with
RCCL all-to-all expert dispatch, a graph-based model-state checkpoint format, and a
provider-driven training loop — with every hot op implemented as a hand-written HIP
kernel for CDNA4 (MFMA/LDS-tiled, wave64) in `d9d_amd/csrc`.
"""

__version__ = "0.1.0"
