"""Building blocks shared by the native models.

GEMMs go through torch.nn.functional.linear (hipBLASLt on ROCm — the
library path for plain GEMMs per the MI355X design rules); everything fused
(norms, rope, attention, activation, sampling) is our HIP kernels via
dynamo_amd.ops.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.nn.functional as F

from dynamo_amd import ops


@dataclass
class AttnMetadata:
    """Everything the attention layers need for one engine step.

    Token order in the flattened batch: decode tokens first (one per decode
    sequence), then prefill chunks back to back.
    """
    slot_mapping: torch.Tensor          # [T] int64
    positions: torch.Tensor             # [T] int32
    # decode part
    num_decode: int = 0
    decode_page_table: Optional[torch.Tensor] = None   # [Bd, maxp] int32
    decode_ctx_lens: Optional[torch.Tensor] = None     # [Bd] int32
    decode_scratch: Optional["ops.DecodeScratch"] = None
    # prefill part
    num_prefill_tokens: int = 0
    prefill_page_table: Optional[torch.Tensor] = None  # [Bp, maxp] int32
    seq_q_start: Optional[torch.Tensor] = None         # [Bp] int32
    seq_q_len: Optional[torch.Tensor] = None
    seq_ctx_len: Optional[torch.Tensor] = None
    prefill_tiles: Optional[tuple] = None
    # sampling
    logits_rows: Optional[torch.Tensor] = None         # [Bs] int64 rows to sample
    # prompt_embeds injection (reference parity: PreprocessedRequest
    # prompt_embeds, lib/llm/src/protocols/common/preprocessor.rs:243):
    # rows of the flattened batch whose input embeddings come from the
    # request instead of the embedding table
    embeds_rows: Optional[torch.Tensor] = None         # [Te] int64
    inputs_embeds: Optional[torch.Tensor] = None       # [Te, hidden]
    # V pages stored d-major (see EngineConfig.kv_v_layout)
    v_transposed: bool = False


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    return F.linear(x, w)


def linear_lora(x: torch.Tensor, w: torch.Tensor, lora, key: str) -> torch.Tensor:
    """F.linear plus the active LoRA low-rank delta, if any."""
    y = F.linear(x, w)
    if lora is not None and key in lora:
        A, B, scale = lora[key]
        y = y + F.linear(F.linear(x, A), B) * scale
    return y


class TPContext:
    """Tensor-parallel context: rank/world + the RCCL process group."""

    def __init__(self, tp_size: int = 1, tp_rank: int = 0, group=None):
        self.size = tp_size
        self.rank = tp_rank
        self.group = group

    def all_reduce(self, x: torch.Tensor) -> torch.Tensor:
        if self.size > 1:
            torch.distributed.all_reduce(x, group=self.group)
        return x


_DEFAULT_TP = TPContext()


def _alloc(shape, device, dtype):
    """Allocate a weight tensor — from the active GMS weight pool when one
    is set (zero-copy shared weights; dynamo_amd.gms), else a fresh tensor.
    Returns (tensor, needs_init)."""
    from dynamo_amd.gms import current_allocator
    pool = current_allocator()
    if pool is not None:
        return pool.allocate(shape, dtype), pool.needs_init
    return torch.empty(shape, device=device, dtype=dtype), True


def init_weight(shape, device, dtype, std=0.02, generator=None):
    from dynamo_amd.gms import current_allocator
    pooled = current_allocator() is not None
    w, needs = _alloc(shape, device, dtype)
    if needs:
        with torch.no_grad():
            if pooled:
                # pool-backed view: generate into a fresh tensor first —
                # normal_ directly into a reinterpreted-uint8 view draws a
                # DIFFERENT sequence on ROCm (HW-observed), which would
                # break GMS build-vs-plain determinism
                tmp = torch.empty(shape, device=device, dtype=dtype)
                tmp.normal_(0.0, std, generator=generator)
                w.copy_(tmp)
            else:
                w.normal_(0.0, std, generator=generator)
    return w


def init_const(shape, device, dtype, val=1.0):
    w, needs = _alloc(shape, device, dtype)
    if needs:
        with torch.no_grad():
            w.fill_(val)
    return w


def init_sharded(shape, device, dtype, tp: "TPContext", dim: int, std=0.02):
    """Generate the FULL weight deterministically, return this rank's shard.

    Every rank consumes the same RNG stream, so shards are consistent and a
    TP-N model computes exactly the same function as TP-1 (up to reduction
    order) — which keeps disagg/TP determinism tests meaningful."""
    if tp.size == 1:
        return init_weight(shape, device, dtype, std)
    n = shape[dim] // tp.size
    shard_shape = list(shape)
    shard_shape[dim] = n
    shard, needs = _alloc(tuple(shard_shape), device, dtype)
    if needs:
        full = torch.empty(shape, device=device, dtype=dtype)
        with torch.no_grad():
            full.normal_(0.0, std)
        shard.copy_(full.narrow(dim, tp.rank * n, n))
        del full
    return shard


class Attention(torch.nn.Module):
    """GQA attention over the paged KV cache (native HIP kernels)."""

    def __init__(self, cfg, layer_idx: int, tp: TPContext, device, dtype):
        super().__init__()
        self.layer_idx = layer_idx
        self.tp = tp
        self.hq = cfg.num_q_heads // tp.size
        self.hkv = max(1, cfg.num_kv_heads // tp.size)
        self.hd = cfg.head_dim
        self.scale = self.hd ** -0.5
        D = cfg.hidden_size
        # fused qkv allocated as ONE tensor (pool-friendly: stays zero-copy
        # under a GMS weight pool); init replays the same RNG stream as
        # separate full q/k/v generation + sharding
        qkv_rows = (self.hq + 2 * self.hkv) * self.hd
        # when tp.size > num_kv_heads, KV heads are REPLICATED: rank r uses
        # kv-head shard (r * num_kv_heads) // tp.size (matches the GQA
        # mapping of the rank's q heads)
        kv_idx = (tp.rank if tp.size <= cfg.num_kv_heads
                  else (tp.rank * cfg.num_kv_heads) // tp.size)
        self.wqkv, needs = _alloc((qkv_rows, D), device, dtype)
        if needs:
            with torch.no_grad():
                off = 0
                for hf, hl, idx in ((cfg.num_q_heads, self.hq, tp.rank),
                                    (cfg.num_kv_heads, self.hkv, kv_idx),
                                    (cfg.num_kv_heads, self.hkv, kv_idx)):
                    full = torch.empty(hf * self.hd, D, device=device,
                                       dtype=dtype).normal_(0.0, 0.02)
                    rows = hl * self.hd
                    self.wqkv[off:off + rows].copy_(
                        full[idx * rows:(idx + 1) * rows]
                        if tp.size > 1 else full)
                    off += rows
        self.wo = init_sharded((D, cfg.num_q_heads * self.hd), device, dtype,
                               tp, 1)
        # qwen2-style QKV bias (same RNG-replay sharding discipline)
        self.bqkv = None
        if getattr(cfg, "attn_bias", False):
            self.bqkv, needs_b = _alloc((qkv_rows,), device, dtype)
            if needs_b:
                with torch.no_grad():
                    off = 0
                    for hf, hl, idx in ((cfg.num_q_heads, self.hq, tp.rank),
                                        (cfg.num_kv_heads, self.hkv, kv_idx),
                                        (cfg.num_kv_heads, self.hkv, kv_idx)):
                        full = torch.empty(hf * self.hd, device=device,
                                           dtype=dtype).normal_(0.0, 0.02)
                        rows = hl * self.hd
                        self.bqkv[off:off + rows].copy_(
                            full[idx * rows:(idx + 1) * rows]
                            if tp.size > 1 else full)
                        off += rows
        self.lora = None  # set by dynamo_amd.lora.LoRAManager

    def forward(self, x, cos_sin, kcache, vcache, meta: AttnMetadata):
        T = x.shape[0]
        qkv = linear_lora(x, self.wqkv, self.lora, "qkv")
        # fused epilogue: strided qkv read (+bias) -> rope -> q contiguous,
        # k/v scattered into the cache (one kernel vs four)
        q = ops.rope_append_qkv(qkv, self.bqkv, meta.positions,
                                meta.slot_mapping, cos_sin, kcache, vcache,
                                self.hq, self.hkv, self.hd,
                                v_transposed=meta.v_transposed)
        qh = q.view(T, self.hq, self.hd)
        out = torch.empty_like(qh)
        nd = meta.num_decode
        if nd:
            ops.paged_attention_decode(
                qh[:nd], kcache, vcache, meta.decode_page_table,
                meta.decode_ctx_lens, self.scale, meta.decode_scratch,
                out=out[:nd], v_transposed=meta.v_transposed)
        if meta.num_prefill_tokens:
            out[nd:] = ops.attention_prefill_paged(
                qh[nd:].contiguous(), kcache, vcache, meta.prefill_page_table,
                meta.seq_q_start, meta.seq_q_len, meta.seq_ctx_len, self.scale,
                meta.prefill_tiles, v_transposed=meta.v_transposed)
        o = linear_lora(out.view(T, self.hq * self.hd), self.wo, self.lora, "o")
        return self.tp.all_reduce(o)


class SwiGLUMLP(torch.nn.Module):
    def __init__(self, cfg, tp: TPContext, device, dtype):
        super().__init__()
        self.tp = tp
        D = cfg.hidden_size
        I = cfg.intermediate_size // tp.size
        self.I = I
        self.w_gate_up, needs = _alloc((2 * I, D), device, dtype)
        if needs:
            with torch.no_grad():
                for sec in range(2):  # gate rows then up rows
                    full = torch.empty(cfg.intermediate_size, D, device=device,
                                       dtype=dtype).normal_(0.0, 0.02)
                    self.w_gate_up[sec * I:(sec + 1) * I].copy_(
                        full[tp.rank * I:(tp.rank + 1) * I]
                        if tp.size > 1 else full)
        self.w_down = init_sharded((D, cfg.intermediate_size), device, dtype,
                                   tp, 1)
        self.lora = None  # set by dynamo_amd.lora.LoRAManager

    def forward(self, x):
        gu = linear_lora(x, self.w_gate_up, self.lora, "gate_up")
        # silu_mul expects [., 2I] with gate then up
        act = ops.silu_mul(gu)
        return self.tp.all_reduce(
            linear_lora(act, self.w_down, self.lora, "down"))
