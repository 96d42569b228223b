"""Op dispatch: native HIP kernels on GPU, torch reference on CPU.

On a GPU box the HIP extension is REQUIRED — a missing/failed extension
raises instead of silently falling back to eager PyTorch, so GPU tests can
never pass on a non-native path.
"""
from __future__ import annotations

import torch

from . import torch_ref
from .torch_ref import make_cos_sin_cache  # re-export

_hip_mod = None
_hip_err: Exception | None = None


def hip():
    """The native kernel extension; raises loudly if unavailable."""
    global _hip_mod, _hip_err
    if _hip_mod is None:
        try:
            from dynamo_amd import _hip as m  # built in-tree by setup.py
            _hip_mod = m
        except Exception as e:  # pragma: no cover
            _hip_err = e
            raise RuntimeError(
                "dynamo_amd._hip native extension is not available on a GPU "
                "device path. Build it with `python setup.py build_ext "
                f"--inplace` (PYTORCH_ROCM_ARCH=gfx950). Original error: {e}"
            ) from e
    return _hip_mod


def hip_available() -> bool:
    try:
        hip()
        return True
    except RuntimeError:
        return False


# --------------------------------------------------------------------------
def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        hip().rmsnorm(out, x.contiguous(), weight, eps)
        return out
    return torch_ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps):
    """residual <- x + residual (in place); returns normalized residual."""
    if x.is_cuda:
        hip().fused_add_rmsnorm(x, residual, weight, eps)
        return x
    return torch_ref.fused_add_rmsnorm(x, residual, weight, eps)


def rope_inplace(q, k, positions, cos_sin, num_q_heads, num_k_heads, head_dim):
    if q.is_cuda:
        hip().rope_inplace(q, k, positions, cos_sin, num_q_heads, num_k_heads,
                           head_dim)
        return q, k
    return torch_ref.rope(q, k, positions, cos_sin, num_q_heads, num_k_heads,
                          head_dim)


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    if gate_up.is_cuda:
        d = gate_up.shape[-1] // 2
        out = torch.empty(gate_up.shape[:-1] + (d,), dtype=gate_up.dtype,
                          device=gate_up.device)
        hip().silu_mul(out, gate_up.contiguous())
        return out
    return torch_ref.silu_mul(gate_up)


def gelu(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        out = torch.empty_like(x)
        hip().gelu(out, x.contiguous())
        return out
    return torch_ref.gelu(x)


def kv_cache_append(kcache, vcache, k, v, slot_mapping,
                    v_transposed=False):
    if kcache.is_cuda:
        hip().kv_cache_append(kcache, vcache, k.contiguous(), v.contiguous(),
                              slot_mapping, v_transposed)
    else:
        torch_ref.kv_cache_append(kcache, vcache, k, v, slot_mapping,
                                  v_transposed)


def rope_append_qkv(qkv, bias, positions, slot_mapping, cos_sin,
                    kcache, vcache, num_q_heads, num_kv_heads, head_dim,
                    v_transposed=False):
    """Fused QKV epilogue: strided qkv read (+bias) -> rope(q,k) ->
    q contiguous out, k/v scattered into the paged cache. One kernel
    instead of {q copy, k copy, rope, kv_append}. Returns q [T, Hq*hd]."""
    T = qkv.shape[0]
    if qkv.is_cuda:
        q_out = torch.empty(T, num_q_heads * head_dim, dtype=qkv.dtype,
                            device=qkv.device)
        hip().rope_append_qkv(q_out, kcache, vcache, qkv, bias, positions,
                              slot_mapping, cos_sin, v_transposed)
        return q_out
    # CPU reference: the unfused sequence
    if bias is not None:
        qkv = qkv + bias
    q, k, v = qkv.split([num_q_heads * head_dim, num_kv_heads * head_dim,
                         num_kv_heads * head_dim], dim=-1)
    q = q.contiguous()
    k = k.contiguous()
    q, k = torch_ref.rope(q, k, positions, cos_sin, num_q_heads,
                          num_kv_heads, head_dim)
    torch_ref.kv_cache_append(kcache, vcache, k.view(T, num_kv_heads, head_dim),
                              v.view(T, num_kv_heads, head_dim), slot_mapping,
                              v_transposed)
    return q


class DecodeScratch:
    """Reusable scratch for two-phase flash-decode (sized once per engine)."""

    def __init__(self, max_batch: int, num_q_heads: int, head_dim: int,
                 max_ctx: int, device):
        self.chunks = int(hip().paged_decode_num_chunks(max_ctx))
        self.chunk_tokens = int(hip().decode_chunk_tokens(max_ctx))
        self.partial = torch.empty(max_batch, num_q_heads, self.chunks, head_dim,
                                   dtype=torch.float32, device=device)
        self.ml = torch.empty(max_batch, num_q_heads, self.chunks, 2,
                              dtype=torch.float32, device=device)
        # fused-merge countdown ([B * Hkv] upper bound; kernel resets it)
        self.chunk_cnt = torch.zeros(max_batch * 128, dtype=torch.int32,
                                     device=device)

    def view(self, batch: int):
        return self.partial[:batch], self.ml[:batch]


def paged_attention_decode(q, kcache, vcache, page_table, ctx_lens, scale,
                           scratch: "DecodeScratch | None" = None, out=None,
                           v_transposed=False):
    if q.is_cuda:
        if out is None:
            out = torch.empty_like(q)
        assert scratch is not None, "GPU decode needs a DecodeScratch"
        partial, ml = scratch.view(q.shape[0])
        hip().paged_attention_decode(out, q, kcache, vcache, page_table,
                                     ctx_lens, partial, ml, scale,
                                     scratch.chunk_tokens, v_transposed,
                                     scratch.chunk_cnt)
        return out
    r = torch_ref.paged_attention_decode(q, kcache, vcache, page_table,
                                         ctx_lens, scale,
                                         v_transposed=v_transposed)
    if out is not None:
        out.copy_(r)
        return out
    return r


def prefill_tile_rows(num_q_heads: int, num_kv_heads: int) -> int:
    """Q-tile granularity of the native prefill kernel: the 8-wave 32x32
    kernel handles GQA groups G in {2,4,8} with (8/G)*32-row tiles; the
    4-wave 16x16 kernel takes 64-row tiles. MUST match the dispatch in
    attention_prefill.hip."""
    if num_q_heads % num_kv_heads == 0:
        g = num_q_heads // num_kv_heads
        if g in (2, 4, 8):
            return (8 // g) * 32
    return 64


def build_prefill_tiles(seq_q_lens, device, rows: int = 64):
    """Host-side tile descriptors for the prefill kernel."""
    tile_seq, tile_q0 = [], []
    for s, ql in enumerate(seq_q_lens):
        for q0 in range(0, int(ql), rows):
            tile_seq.append(s)
            tile_q0.append(q0)
    return (torch.tensor(tile_seq, dtype=torch.int32, device=device),
            torch.tensor(tile_q0, dtype=torch.int32, device=device))


def attention_prefill_paged(q, kcache, vcache, page_table, seq_q_start,
                            seq_q_len, seq_ctx_len, scale, tiles=None,
                            v_transposed=False):
    if q.is_cuda:
        out = torch.empty_like(q)
        if tiles is None:
            rows = prefill_tile_rows(q.shape[1], kcache.shape[1])
            tiles = build_prefill_tiles(seq_q_len.tolist(), q.device, rows)
        tile_seq, tile_q0 = tiles
        hip().attention_prefill_paged(out, q, kcache, vcache, page_table,
                                      tile_seq, tile_q0, seq_q_start,
                                      seq_q_len, seq_ctx_len, scale,
                                      v_transposed)
        return out
    return torch_ref.attention_prefill_paged(q, kcache, vcache, page_table,
                                             seq_q_start, seq_q_len,
                                             seq_ctx_len, scale,
                                             v_transposed=v_transposed)


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.int32,
                          device=logits.device)
        hip().greedy_sample(out, logits.float().contiguous())
        return out
    return torch_ref.greedy_sample(logits)


def _splitmix64(x: torch.Tensor) -> torch.Tensor:
    """splitmix64 finalizer on int64 tensors (wrapping arithmetic; torch
    >> is arithmetic, so logical shifts are emulated with masks). Bit-for-
    bit the same hash as the HIP sampling kernel's hash_u64."""
    def lsr(v, k):
        return (v >> k) & ((1 << (64 - k)) - 1)
    x = x + 0x9e3779b97f4a7c15
    x = (x ^ lsr(x, 30)) * -0x40a7b892e31b1a47   # 0xbf58476d1ce4e5b9
    x = (x ^ lsr(x, 27)) * -0x6b2fb644ecceee15   # 0x94d049bb133111eb
    return x ^ lsr(x, 31)


def gumbel_sample(logits: torch.Tensor, inv_temp: torch.Tensor,
                  seed: int, row_seeds: torch.Tensor = None) -> torch.Tensor:
    if logits.is_cuda:
        out = torch.empty(logits.shape[0], dtype=torch.int32,
                          device=logits.device)
        hip().gumbel_sample(out, logits.float().contiguous(), inv_temp, seed,
                            row_seeds)
        return out
    # CPU: DETERMINISTIC counter-based Gumbel noise — the same
    # splitmix64(row_seed ^ v) stream as the HIP kernel (row_seed defaults
    # to seed ^ b<<32), so sampling is reproducible across
    # engines/processes (TP lockstep needs this)
    B, V = logits.shape
    if row_seeds is None:
        rs = (torch.tensor(seed, dtype=torch.int64)
              ^ (torch.arange(B, dtype=torch.int64) << 32))
    else:
        rs = row_seeds.to(torch.int64).cpu()
    v = torch.arange(V, dtype=torch.int64).unsqueeze(0)
    h = _splitmix64(rs.unsqueeze(1) ^ v)
    u = (((h >> 11) & ((1 << 53) - 1)) + 1).double() * (2.0 ** -53)
    g = -torch.log(-torch.log(u.float()))
    return (logits.float() * inv_temp.unsqueeze(-1) + g).argmax(-1).to(torch.int32)


def topkp_sample(logits: torch.Tensor, inv_temp: torch.Tensor,
                 top_k: torch.Tensor, top_p: torch.Tensor,
                 seed: int, row_seeds: torch.Tensor = None) -> torch.Tensor:
    """Fused top-k/top-p Gumbel sampling (GPU only; per-row params)."""
    out = torch.empty(logits.shape[0], dtype=torch.int32,
                      device=logits.device)
    hip().topkp_sample(out, logits.float().contiguous(), inv_temp,
                       top_k, top_p, seed, row_seeds)
    return out


def topk_gating(logits: torch.Tensor, k: int):
    """softmax + renormalized top-k over experts. logits [T, E] fp32."""
    if logits.is_cuda:
        T = logits.shape[0]
        topw = torch.empty(T, k, dtype=torch.float32, device=logits.device)
        topi = torch.empty(T, k, dtype=torch.int32, device=logits.device)
        hip().topk_gating(topw, topi, logits.float().contiguous())
        return topw, topi
    p = torch.softmax(logits.float(), dim=-1)
    topw, topi = torch.topk(p, k, dim=-1)
    return topw / topw.sum(-1, keepdim=True), topi.to(torch.int32)


def build_moe_tiles(counts, max_m: int = 16):
    """Host-side tile list [(expert, row0, m)] for the grouped GEMM;
    segments larger than max_m split into multiple tiles."""
    tiles = []
    r0 = 0
    for e, n in enumerate(counts):
        off = 0
        while off < n:
            m = min(max_m, n - off)
            tiles.append((e, r0 + off, m))
            off += m
        r0 += n
    return tiles


def moe_grouped_gemm(x: torch.Tensor, w: torch.Tensor, tiles_t: torch.Tensor):
    """x [T, D] gathered by expert; w [E, N, D]; -> y [T, N]."""
    y = torch.empty(x.shape[0], w.shape[1], dtype=x.dtype, device=x.device)
    hip().moe_grouped_gemm(y, x.contiguous(), w, tiles_t)
    return y


def moe_grouped_gemm_seg(x, w, seg_start, max_tokens):
    """Sync-free grouped GEMM: seg_start [E+1] int32 on DEVICE."""
    y = torch.empty(x.shape[0], w.shape[1], dtype=x.dtype, device=x.device)
    hip().moe_grouped_gemm_seg(y, x.contiguous(), w, seg_start, max_tokens)
    return y


def gather_pages(staging, cache, page_ids):
    if cache.is_cuda:
        hip().gather_pages(staging, cache, page_ids)
    else:
        n = page_ids.shape[0]
        pe = cache[0].numel()
        staging.view(-1)[: n * pe] = cache[page_ids.long()].reshape(-1)


def scatter_pages(staging, cache, page_ids):
    if cache.is_cuda:
        hip().scatter_pages(staging, cache, page_ids)
    else:
        n = page_ids.shape[0]
        pe = cache[0].numel()
        cache[page_ids.long()] = staging.view(-1)[: n * pe].view(n, *cache.shape[1:])


def copy_pages(dst_cache, src_cache, pairs):
    if src_cache.is_cuda:
        hip().copy_pages(dst_cache, src_cache, pairs)
    else:
        src = pairs[:, 0].long()
        dst = pairs[:, 1].long()
        dst_cache[dst] = src_cache[src]
