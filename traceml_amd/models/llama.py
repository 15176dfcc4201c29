"""Llama-3-8B workload (BASELINE config 4): random-init via HF transformers'
LlamaConfig/LlamaForCausalLM (no network, no checkpoint). ``tiny=True``
gives a CPU-testable miniature with the same architecture family (RMSNorm,
SwiGLU, RoPE, GQA)."""

from __future__ import annotations


def build_llama3(tiny: bool = False, seq_len: int = 8192, device=None):
    """device="cuda" initializes weights directly on the GPU (an 8B random
    init on CPU costs minutes; on-device it is seconds)."""
    import torch
    from transformers import LlamaConfig, LlamaForCausalLM

    if tiny:
        config = LlamaConfig(
            vocab_size=512,
            hidden_size=64,
            intermediate_size=128,
            num_hidden_layers=2,
            num_attention_heads=4,
            num_key_value_heads=2,
            max_position_embeddings=256,
        )
    else:
        # Llama-3-8B architecture (public config shape)
        config = LlamaConfig(
            vocab_size=128256,
            hidden_size=4096,
            intermediate_size=14336,
            num_hidden_layers=32,
            num_attention_heads=32,
            num_key_value_heads=8,
            max_position_embeddings=seq_len,
            rope_theta=500000.0,
        )
    config.use_cache = False
    if device is not None:
        with torch.device(device):
            return LlamaForCausalLM(config)
    return LlamaForCausalLM(config)
