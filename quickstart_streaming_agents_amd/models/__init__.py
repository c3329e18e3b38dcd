"""Model factory: named presets -> decoder instances.

BASELINE.json configs name two decoder families: Llama-3-8B (config 3, the
flagship agent LLM) and Mixtral-8x7B MoE (config 5).
"""

from __future__ import annotations


def build_model(name: str, device: str = "cuda", dtype=None, seed: int = 0,
                tp_rank: int = 0, tp_size: int = 1, tp_group=None,
                ep_size: int | None = None):
    import torch
    dtype = dtype or (torch.bfloat16 if str(device).startswith("cuda")
                      else torch.float32)
    if name.startswith("mixtral") or name == "tiny-moe":
        from .mixtral import MixtralConfig, MixtralModel
        return MixtralModel(MixtralConfig.preset(name), device=device,
                            dtype=dtype, seed=seed, tp_rank=tp_rank,
                            tp_size=tp_size, tp_group=tp_group,
                            ep_size=ep_size)
    from .llama import LlamaConfig, LlamaModel
    return LlamaModel(LlamaConfig.preset(name), device=device, dtype=dtype,
                      seed=seed, tp_rank=tp_rank, tp_size=tp_size,
                      tp_group=tp_group)
