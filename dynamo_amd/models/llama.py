"""Llama-family causal LM (Llama-3 8B/70B, Qwen2-style also fits) —
MI355X-native forward over the paged KV cache.

Weights are random-initialized by default (BASELINE.md benches on
synthetic data / random weights); local HF safetensors checkpoints load
via models/loader.py (`--model /path/to/checkpoint`).
"""
from __future__ import annotations

import torch

from dynamo_amd import ops
from dynamo_amd.ops import torch_ref
from .layers import (AttnMetadata, Attention, SwiGLUMLP, TPContext,
                     init_const, init_weight, linear)


class LlamaDecoderLayer(torch.nn.Module):
    def __init__(self, cfg, layer_idx, tp, device, dtype):
        super().__init__()
        self.attn = Attention(cfg, layer_idx, tp, device, dtype)
        self.mlp = SwiGLUMLP(cfg, tp, device, dtype)
        self.input_norm_w = init_const((cfg.hidden_size,), device, dtype, 1.0)
        self.post_norm_w = init_const((cfg.hidden_size,), device, dtype, 1.0)
        self.eps = cfg.rms_eps

    def forward(self, x, residual, cos_sin, kcache, vcache, meta):
        if residual is None:
            residual = x.clone()
            x = ops.rmsnorm(x, self.input_norm_w, self.eps)
        else:
            x = ops.fused_add_rmsnorm(x, residual, self.input_norm_w, self.eps)
        x = self.attn.forward(x, cos_sin, kcache, vcache, meta)
        x = ops.fused_add_rmsnorm(x, residual, self.post_norm_w, self.eps)
        x = self.mlp.forward(x)
        return x, residual


class LlamaForCausalLM(torch.nn.Module):
    def __init__(self, cfg, device="cpu", dtype=torch.bfloat16,
                 tp: TPContext | None = None, seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.tp = tp or TPContext()
        torch.manual_seed(seed)  # same weights on every TP rank pre-shard
        self.embed = init_weight((cfg.vocab_size, cfg.hidden_size), device, dtype)
        self.layers = torch.nn.ModuleList([
            LlamaDecoderLayer(cfg, i, self.tp, device, dtype)
            for i in range(cfg.num_layers)
        ])
        self.final_norm_w = init_const((cfg.hidden_size,), device, dtype, 1.0)
        self.lm_head = (self.embed if cfg.tie_embeddings
                        else init_weight((cfg.vocab_size, cfg.hidden_size),
                                         device, dtype))
        self.cos_sin = torch_ref.make_cos_sin_cache(
            cfg.max_position, cfg.head_dim, cfg.rope_theta, device=device,
            scaling=cfg.rope_scaling)

    def forward(self, input_ids, kv_pool, meta: AttnMetadata):
        x = torch.nn.functional.embedding(input_ids.long(), self.embed)
        if meta.inputs_embeds is not None:
            x = x.index_copy(0, meta.embeds_rows,
                             meta.inputs_embeds.to(x.dtype))
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer.forward(x, residual, self.cos_sin,
                                        kv_pool.kcache(i), kv_pool.vcache(i),
                                        meta)
        # final residual add + norm
        x = ops.fused_add_rmsnorm(x, residual, self.final_norm_w, self.cfg.rms_eps)
        return x

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        return linear(hidden, self.lm_head).float()
