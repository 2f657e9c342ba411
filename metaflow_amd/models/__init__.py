"""Model zoo for the @parallel train-step hot path: Llama-3 dense and
Mixtral MoE, built directly on metaflow_amd.ops HIP kernels (attention,
RMSNorm, RoPE, SwiGLU, fused CE) with hipBLASLt for the plain GEMMs."""

from .llama import LlamaConfig, LlamaForCausalLM

__all__ = ["LlamaConfig", "LlamaForCausalLM"]
