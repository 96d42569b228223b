"""OPT-family causal LM (config #1: OPT-125m aggregated on CPU).

Architecture: learned positional embeddings, pre-LayerNorm, MHA, GELU MLP,
tied LM head. Runs through the same paged-KV engine path as Llama; on CPU it
uses the torch reference ops (the native GPU kernels target head_dim=128
models — OPT is the CPU plumbing config from BASELINE.md)."""
from __future__ import annotations

import torch
import torch.nn.functional as F

from dynamo_amd import ops
from .layers import (AttnMetadata, Attention, TPContext, init_const,
                     init_weight, linear)


class OPTDecoderLayer(torch.nn.Module):
    def __init__(self, cfg, layer_idx, tp, device, dtype):
        super().__init__()
        D = cfg.hidden_size
        self.attn = Attention(cfg, layer_idx, tp, device, dtype)
        self.ln1_w = init_const((D,), device, dtype, 1.0)
        self.ln1_b = init_const((D,), device, dtype, 0.0)
        self.ln2_w = init_const((D,), device, dtype, 1.0)
        self.ln2_b = init_const((D,), device, dtype, 0.0)
        self.fc1 = init_weight((cfg.intermediate_size, D), device, dtype)
        self.fc2 = init_weight((D, cfg.intermediate_size), device, dtype)

    def _ln(self, x, w, b):
        return F.layer_norm(x.float(), (x.shape[-1],), w.float(), b.float()).to(x.dtype)

    def forward(self, x, cos_sin, kcache, vcache, meta):
        h = self._ln(x, self.ln1_w, self.ln1_b)
        x = x + self.attn.forward(h, cos_sin, kcache, vcache, meta)
        h = self._ln(x, self.ln2_w, self.ln2_b)
        h = ops.gelu(linear(h, self.fc1))
        x = x + linear(h, self.fc2)
        return x


class OPTForCausalLM(torch.nn.Module):
    def __init__(self, cfg, device="cpu", dtype=torch.bfloat16,
                 tp: TPContext | None = None, seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.tp = tp or TPContext()
        torch.manual_seed(seed)
        self.embed = init_weight((cfg.vocab_size, cfg.hidden_size), device, dtype)
        self.pos_embed = init_weight((cfg.max_position, cfg.hidden_size),
                                     device, dtype)
        self.layers = torch.nn.ModuleList([
            OPTDecoderLayer(cfg, i, self.tp, device, dtype)
            for i in range(cfg.num_layers)
        ])
        self.final_ln_w = init_const((cfg.hidden_size,), device, dtype, 1.0)
        self.final_ln_b = init_const((cfg.hidden_size,), device, dtype, 0.0)
        self.lm_head = self.embed  # tied
        # OPT has no rotary cache; Attention.forward still expects one.
        # A zero-rotation table (cos=1, sin=0) makes rope a no-op.
        half = cfg.head_dim // 2
        cs = torch.zeros(cfg.max_position, cfg.head_dim, dtype=torch.float32,
                         device=device)
        cs[:, :half] = 1.0
        self.cos_sin = cs

    def forward(self, input_ids, kv_pool, meta: AttnMetadata):
        x = F.embedding(input_ids.long(), self.embed)
        if meta.inputs_embeds is not None:
            x = x.index_copy(0, meta.embeds_rows,
                             meta.inputs_embeds.to(x.dtype))
        x = x + F.embedding(meta.positions.long(), self.pos_embed)
        for i, layer in enumerate(self.layers):
            x = layer.forward(x, self.cos_sin, kv_pool.kcache(i),
                              kv_pool.vcache(i), meta)
        x = F.layer_norm(x.float(), (x.shape[-1],), self.final_ln_w.float(),
                         self.final_ln_b.float()).to(x.dtype)
        return x

    def compute_logits(self, hidden):
        return linear(hidden, self.lm_head).float()
