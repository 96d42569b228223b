"""Batch sampling: greedy (native HIP argmax) and temperature sampling via
the native Gumbel-max kernel; top-k/top-p apply a torch-side logit filter
first (filtered Gumbel-max samples the renormalized distribution exactly)."""
from __future__ import annotations

from typing import List

import torch

from dynamo_amd import ops
from .scheduler import Request


def _filter_topk_topp(logits: torch.Tensor, top_k: int, top_p: float):
    if top_k > 0 and top_k < logits.shape[-1]:
        kth = torch.topk(logits, top_k, dim=-1).values[..., -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    if top_p < 1.0:
        sorted_logits, idx = torch.sort(logits, descending=True, dim=-1)
        probs = torch.softmax(sorted_logits, dim=-1)
        cum = probs.cumsum(-1)
        # keep tokens until cumulative prob exceeds top_p (always keep first)
        drop = cum - probs > top_p
        sorted_logits = sorted_logits.masked_fill(drop, float("-inf"))
        logits = torch.full_like(logits, float("-inf")).scatter(
            -1, idx, sorted_logits)
    return logits


def compute_logprobs(logits: torch.Tensor, sampled: torch.Tensor,
                     reqs: List[Request]):
    """Per-request top-N logprobs (None for requests with logprobs=0).

    Returns a list aligned with reqs: each entry is
    {"token_logprob": float, "top": [[token_id, logprob], ...]} computed
    from log_softmax of the full-vocab logits row."""
    want = [r.sampling.logprobs for r in reqs]
    if not any(want):
        return [None] * len(reqs)
    lp = torch.log_softmax(logits.float(), dim=-1)
    kmax = min(max(want), lp.shape[-1])
    topv, topi = torch.topk(lp, kmax, dim=-1)
    own = lp.gather(-1, sampled.long().unsqueeze(-1)).squeeze(-1)
    topv_l, topi_l, own_l = topv.tolist(), topi.tolist(), own.tolist()
    out = []
    for i, n in enumerate(want):
        if not n:
            out.append(None)
            continue
        out.append({"token_logprob": own_l[i],
                    "top": [[topi_l[i][j], topv_l[i][j]]
                            for j in range(min(n, kmax))]})
    return out


_GOLDEN = 0x9E3779B97F4A7C15


def _row_seeds(reqs: List[Request], step_seed: int,
               device) -> torch.Tensor | None:
    """Per-row RNG seeds honoring client-supplied sampling seeds.

    Seeded requests get a stream keyed by (user_seed, output position) —
    reproducible regardless of batching/scheduling; unseeded rows keep the
    engine-step stream (seed ^ b<<32, bit-identical to the scalar path).
    Returns None when no request carries a seed (scalar fast path)."""
    if not any(r.sampling.seed for r in reqs):
        return None
    mask = (1 << 64) - 1
    rows = []
    for b, r in enumerate(reqs):
        if r.sampling.seed:
            rows.append((r.sampling.seed * _GOLDEN
                         + (len(r.output_tokens) + 1)) & mask)
        else:
            rows.append((step_seed ^ (b << 32)) & mask)
    t = torch.tensor([x - (1 << 64) if x >= (1 << 63) else x
                      for x in rows], dtype=torch.int64)
    return t.to(device) if device.type == "cuda" else t


def sample_tokens(logits: torch.Tensor, reqs: List[Request],
                  step_seed: int) -> torch.Tensor:
    """logits [n, V] fp32 -> token ids [n] int32."""
    n = logits.shape[0]
    assert n == len(reqs)
    temps = [r.sampling.temperature for r in reqs]
    if all(t == 0.0 for t in temps):
        return ops.greedy_sample(logits)

    row_seeds = _row_seeds(reqs, step_seed, logits.device)
    greedy_mask = torch.tensor([t == 0.0 for t in temps], device=logits.device)
    inv_t = torch.tensor([1.0 / t if t > 0 else 1.0 for t in temps],
                         dtype=torch.float32, device=logits.device)
    has_filter = any(r.sampling.top_k > 0 or r.sampling.top_p < 1.0
                     for r in reqs)
    if has_filter and logits.is_cuda:
        # native fused kernel: per-row thresholds, no sort, no host sync
        tk = torch.tensor([r.sampling.top_k for r in reqs],
                          dtype=torch.int32, device=logits.device)
        tp = torch.tensor([r.sampling.top_p for r in reqs],
                          dtype=torch.float32, device=logits.device)
        sampled = ops.topkp_sample(logits, inv_t, tk, tp, step_seed,
                                   row_seeds)
    else:
        filt = logits
        if has_filter:
            # CPU: per-request torch filters, then filtered Gumbel-max
            rows = []
            for i, r in enumerate(reqs):
                rows.append(_filter_topk_topp(logits[i:i + 1],
                                              r.sampling.top_k,
                                              r.sampling.top_p))
            filt = torch.cat(rows, 0)
        sampled = ops.gumbel_sample(filt, inv_t, step_seed, row_seeds)
    if greedy_mask.any():
        greedy = ops.greedy_sample(logits)
        sampled = torch.where(greedy_mask, greedy, sampled)
    return sampled
