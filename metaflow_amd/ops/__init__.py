"""metaflow_amd.ops — hand-written CDNA4 HIP kernels for the training hot
path (RMSNorm, RoPE, fused Adam, flash attention, SwiGLU, cross entropy)
plus the native C++ CAS engine.

Kernels are compiled for gfx950 only (no multi-backend dispatch). On a GPU
box, requesting an op whose extension failed to load raises
KernelExtensionMissing — there is no silent eager fallback on the GPU path.
"""
