"""Plain-PyTorch fp32 reference implementations of every native op.

These serve two roles:
  1. the CPU execution path (config #1: OPT-125m aggregated on CPU), and
  2. the numerics oracle the HIP kernels are tested against
     (tests/test_gpu_kernels.py compares each kernel to these in fp32).
"""
from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    return ((xf * torch.rsqrt(var + eps)) * weight.float()).to(x.dtype)


def fused_add_rmsnorm(x, residual, weight, eps):
    """residual <- x + residual ; returns rmsnorm(residual) (matches the HIP
    kernel's bf16 residual-stream rounding: the sum is stored in the stream
    dtype before normalizing)."""
    summed = (x.float() + residual.float()).to(residual.dtype)
    residual.copy_(summed)
    return rmsnorm(summed, weight, eps)


def make_cos_sin_cache(max_pos: int, head_dim: int, theta: float,
                       device="cpu", scaling=None) -> torch.Tensor:
    """[max_pos, head_dim] fp32: first half cos, second half sin."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64) / half))
    if scaling is not None:
        inv_freq = apply_llama3_scaling(inv_freq, **scaling)
    t = torch.arange(max_pos, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float().to(device)


def apply_llama3_scaling(inv_freq, factor=8.0, low_freq_factor=1.0,
                         high_freq_factor=4.0, original_max_position=8192):
    import math
    low_wavelen = original_max_position / low_freq_factor
    high_wavelen = original_max_position / high_freq_factor
    new = []
    for f in inv_freq:
        wavelen = 2 * math.pi / f
        if wavelen < high_wavelen:
            new.append(f)
        elif wavelen > low_wavelen:
            new.append(f / factor)
        else:
            smooth = (original_max_position / wavelen - low_freq_factor) / (
                high_freq_factor - low_freq_factor)
            new.append((1 - smooth) * f / factor + smooth * f)
    return torch.tensor(new, dtype=inv_freq.dtype)


def rope(q, k, positions, cos_sin, num_q_heads, num_k_heads, head_dim):
    """NeoX-style rotary, in-place semantics (returns new tensors)."""
    half = head_dim // 2
    cs = cos_sin[positions.long()]          # [T, hd]
    cos = cs[:, :half].unsqueeze(1).float()  # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1).float()

    def rot(x, H):
        x = x.view(-1, H, head_dim).float()
        x1, x2 = x[..., :half], x[..., half:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1)

    qo = rot(q, num_q_heads).to(q.dtype).view(q.shape)
    ko = rot(k, num_k_heads).to(k.dtype).view(k.shape)
    return qo, ko


def silu_mul(gate_up: torch.Tensor) -> torch.Tensor:
    d = gate_up.shape[-1] // 2
    g, u = gate_up[..., :d].float(), gate_up[..., d:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def gelu(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x.float(), approximate="tanh").to(x.dtype)


def kv_cache_append(kcache, vcache, k, v, slot_mapping,
                    v_transposed=False):
    """kcache: [P, Hkv, ps, hd]; k/v: [T, Hkv, hd]; slots: [T] int64.
    v_transposed: vcache is d-major [P, Hkv, hd, ps] (the MI355X decode
    layout — PV fragments read as contiguous token runs)."""
    P, Hkv, ps, hd = kcache.shape
    valid = slot_mapping >= 0
    slots = slot_mapping[valid]
    pages = (slots // ps).long()
    offs = (slots % ps).long()
    kcache[pages, :, offs] = k[valid].to(kcache.dtype)
    if v_transposed:
        vcache[pages, :, :, offs] = v[valid].to(vcache.dtype)
    else:
        vcache[pages, :, offs] = v[valid].to(vcache.dtype)


def _gather_kv(kcache, page_table, ctx_len, v_transposed=False):
    """-> [ctx_len, Hkv, hd] for one sequence."""
    if v_transposed:
        P, Hkv, hd, ps = kcache.shape
    else:
        P, Hkv, ps, hd = kcache.shape
    npages = (ctx_len + ps - 1) // ps
    pages = page_table[:npages].long()
    kv = kcache[pages]
    if v_transposed:                        # [np, Hkv, hd, ps]
        kv = kv.permute(0, 3, 1, 2).reshape(npages * ps, Hkv, hd)
    else:                                   # [np, Hkv, ps, hd]
        kv = kv.permute(0, 2, 1, 3).reshape(npages * ps, Hkv, hd)
    return kv[:ctx_len]


def paged_attention_decode(q, kcache, vcache, page_table, ctx_lens, scale,
                           v_transposed=False):
    """q: [B, Hq, hd] -> out [B, Hq, hd]."""
    B, Hq, hd = q.shape
    Hkv = kcache.shape[1]
    G = Hq // Hkv
    outs = []
    for b in range(B):
        ctx = int(ctx_lens[b])
        kk = _gather_kv(kcache, page_table[b], ctx).float()  # [ctx, Hkv, hd]
        vv = _gather_kv(vcache, page_table[b], ctx, v_transposed).float()
        qq = q[b].float()                                    # [Hq, hd]
        kk = kk.repeat_interleave(G, dim=1)                  # [ctx, Hq, hd]
        vv = vv.repeat_interleave(G, dim=1)
        s = torch.einsum("hd,thd->ht", qq, kk) * scale       # [Hq, ctx]
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("ht,thd->hd", p, vv)
        outs.append(o)
    return torch.stack(outs).to(q.dtype)


def attention_prefill_paged(q, kcache, vcache, page_table, seq_q_start,
                            seq_q_len, seq_ctx_len, scale,
                            v_transposed=False):
    """Varlen causal attention where K/V come from the paged cache.

    q: [Tq, Hq, hd]; query token j of seq s sits at absolute position
    ctx_len - q_len + j and attends kv positions <= its own.
    """
    Tq, Hq, hd = q.shape
    Hkv = kcache.shape[1]
    G = Hq // Hkv
    out = torch.zeros_like(q)
    nseq = len(seq_q_len)
    for s in range(nseq):
        qs, ql, ctx = int(seq_q_start[s]), int(seq_q_len[s]), int(seq_ctx_len[s])
        if ql == 0:
            continue
        kk = _gather_kv(kcache, page_table[s], ctx).float()
        vv = _gather_kv(vcache, page_table[s], ctx, v_transposed).float()
        qq = q[qs:qs + ql].float()                           # [ql, Hq, hd]
        kk = kk.repeat_interleave(G, dim=1)
        vv = vv.repeat_interleave(G, dim=1)
        sc = torch.einsum("qhd,thd->hqt", qq, kk) * scale    # [Hq, ql, ctx]
        qpos = torch.arange(ctx - ql, ctx, device=q.device)
        tpos = torch.arange(ctx, device=q.device)
        mask = tpos[None, :] > qpos[:, None]                 # [ql, ctx]
        sc = sc.masked_fill(mask[None], float("-inf"))
        p = torch.softmax(sc, dim=-1)
        o = torch.einsum("hqt,thd->qhd", p, vv)
        out[qs:qs + ql] = o.to(q.dtype)
    return out


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    return logits.argmax(dim=-1).to(torch.int32)
