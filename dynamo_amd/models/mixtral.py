"""Mixtral-style MoE causal LM (config #5).

Top-k gating + expert MLPs. The expert compute is segment-batched per
expert (sorted token dispatch); expert-parallel all-to-all over RCCL is
layered in the parallel package. A HIP grouped-GEMM fast path replaces the
per-expert loop when available (dynamo_amd.ops.moe)."""
from __future__ import annotations

import torch
import torch.nn.functional as F

from dynamo_amd import ops
from .layers import (AttnMetadata, Attention, TPContext, init_const,
                     init_weight, linear)


class MoEMLP(torch.nn.Module):
    """Two parallelism modes over the TP group (reference parity:
    planner/config/parallelization.py moe_tp_size / moe_ep_size):
      MoE-TP (default): every rank holds all experts with the intermediate
        dim sharded; all-reduce after down-proj.
      EP (cfg.model.moe_ep): experts are SHARDED across ranks at full
        intermediate width; every rank computes its local experts for the
        whole batch and the same all-reduce sums the partial outputs —
        each (token, expert) pair is computed on exactly one rank."""

    def __init__(self, cfg, tp: TPContext, device, dtype):
        super().__init__()
        self.tp = tp
        self.E = cfg.num_experts
        self.topk = cfg.num_experts_per_tok
        self.ep = cfg.moe_ep and tp.size > 1
        D = cfg.hidden_size
        from .layers import _alloc, init_sharded
        self.router = init_weight((self.E, D), device, dtype)
        if self.ep:
            assert self.E % tp.size == 0, "num_experts % ep_size != 0"
            El = self.E // tp.size
            self.e0 = tp.rank * El
            self.El = El
            I = cfg.intermediate_size
            self.I = I
            self.w_gate_up, needs = _alloc((El, 2 * I, D), device, dtype)
            self.w_down, needs2 = _alloc((El, D, I), device, dtype)
            if needs:
                with torch.no_grad():
                    for sec in range(2):  # gate then up (same RNG order)
                        full = torch.empty(self.E, I, D, device=device,
                                           dtype=dtype).normal_(0.0, 0.02)
                        self.w_gate_up[:, sec * I:(sec + 1) * I].copy_(
                            full[self.e0:self.e0 + El])
            if needs2:
                with torch.no_grad():
                    full = torch.empty(self.E, D, I, device=device,
                                       dtype=dtype).normal_(0.0, 0.02)
                    self.w_down.copy_(full[self.e0:self.e0 + El])
            return
        self.e0, self.El = 0, self.E
        I = cfg.intermediate_size // tp.size
        self.I = I
        # fused per-expert weights [E, 2I_local, D], one pool allocation
        self.w_gate_up, needs = _alloc((self.E, 2 * I, D), device, dtype)
        if needs:
            with torch.no_grad():
                for sec in range(2):  # gate then up
                    full = torch.empty(self.E, cfg.intermediate_size, D,
                                       device=device, dtype=dtype).normal_(0.0, 0.02)
                    self.w_gate_up[:, sec * I:(sec + 1) * I].copy_(
                        full[:, tp.rank * I:(tp.rank + 1) * I]
                        if tp.size > 1 else full)
        self.w_down = init_sharded((self.E, D, cfg.intermediate_size), device,
                                   dtype, tp, 2)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        T, D = x.shape
        logits = linear(x, self.router).float()            # [T, E]
        topw, topi = ops.topk_gating(logits, self.topk)    # [T, k]
        topw = topw.to(x.dtype)

        out = torch.zeros_like(x)
        flat_expert = topi.reshape(-1).long()              # [T*k]
        flat_token = (torch.arange(T, device=x.device)
                      .repeat_interleave(self.topk))       # [T*k]
        # sort by expert -> contiguous segments
        order = torch.argsort(flat_expert, stable=True)
        seg_expert = flat_expert[order]
        seg_token = flat_token[order]
        # scatter_add instead of bincount (bincount is not hipGraph-capturable)
        counts = torch.zeros(self.E, dtype=torch.long, device=x.device)
        counts.scatter_add_(0, seg_expert,
                            torch.ones_like(seg_expert, dtype=torch.long))
        if x.is_cuda and T <= 256:
            # decode regime: fully device-side segment offsets — sync-free
            # and hipGraph-capturable. EP ranks pass their expert window's
            # offsets; the gathered rows outside the window are never
            # touched by the kernel (their yg rows stay undefined and are
            # excluded from the index_add below via the host EP slice).
            seg_start_full = torch.zeros(self.E + 1, dtype=torch.int32,
                                         device=x.device)
            seg_start_full[1:] = torch.cumsum(counts, 0).to(torch.int32)
            if not self.ep:
                import os as _os
                use_bmm = (_os.environ.get("DYNAMO_MOE_BMM", "1") != "0"
                           and T <= 64)
                if use_bmm:
                    # capacity-padded strided-batched GEMM: pad each
                    # expert's tokens to cap=T rows and run ONE hipBLASLt
                    # batched GEMM per projection — every expert's W
                    # streams exactly once at dense-GEMM rates (~6 TB/s vs
                    # ~3 for the grouped kernel). The E/topk flop padding
                    # is free in the W-bandwidth-bound decode regime.
                    # Fully device-side + static shapes: hipGraph-safe.
                    cap = T
                    slot = (torch.arange(T * self.topk, device=x.device)
                            - seg_start_full[seg_expert].long())
                    pad_idx = seg_expert * cap + slot
                    xpad = torch.zeros(self.E * cap, D, dtype=x.dtype,
                                       device=x.device)
                    xpad[pad_idx] = x[seg_token]
                    gu = torch.bmm(xpad.view(self.E, cap, D),
                                   self.w_gate_up.transpose(1, 2))
                    act = ops.silu_mul(gu.reshape(self.E * cap, -1))
                    ypad = torch.bmm(act.view(self.E, cap, -1),
                                     self.w_down.transpose(1, 2))
                    yg = ypad.reshape(self.E * cap, D)[pad_idx]
                else:
                    xg = x[seg_token]
                    gu = ops.moe_grouped_gemm_seg(
                        xg, self.w_gate_up, seg_start_full, T * self.topk)
                    act = ops.silu_mul(gu)
                    yg = ops.moe_grouped_gemm_seg(
                        act, self.w_down, seg_start_full, T * self.topk)
                w = topw.reshape(-1)[order].unsqueeze(-1)
                out.index_add_(0, seg_token, (yg * w).to(x.dtype))
                return self.tp.all_reduce(out)
        # host-side paths: EP slicing and/or large prefill segments
        counts_l = counts.tolist()
        if self.ep:
            lo = sum(counts_l[:self.e0])
            hi = lo + sum(counts_l[self.e0:self.e0 + self.El])
            order = order[lo:hi]
            seg_token = seg_token[lo:hi]
            counts_l = counts_l[self.e0:self.e0 + self.El]
        xg = x[seg_token]                                  # [local, D]
        if x.is_cuda and T <= 256:
            tiles = ops.build_moe_tiles(counts_l)
            tiles_t = torch.tensor(tiles, dtype=torch.int32,
                                   device=x.device).view(-1, 3)
            gu = ops.moe_grouped_gemm(xg, self.w_gate_up, tiles_t)
            act = ops.silu_mul(gu)
            yg = ops.moe_grouped_gemm(act, self.w_down, tiles_t)
        else:
            # prefill regime: large segments -> hipBLASLt per expert
            yg = torch.empty_like(xg)
            s = 0
            for e in range(len(counts_l)):
                n = counts_l[e]
                if n == 0:
                    continue
                gu = linear(xg[s:s + n], self.w_gate_up[e])
                yg[s:s + n] = linear(ops.silu_mul(gu), self.w_down[e])
                s += n
        w = topw.reshape(-1)[order].unsqueeze(-1)
        out.index_add_(0, seg_token, (yg * w).to(x.dtype))
        return self.tp.all_reduce(out)


class MixtralDecoderLayer(torch.nn.Module):
    def __init__(self, cfg, layer_idx, tp, device, dtype):
        super().__init__()
        self.attn = Attention(cfg, layer_idx, tp, device, dtype)
        self.moe = MoEMLP(cfg, tp, device, dtype)
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
        x = self.moe.forward(x)
        return x, residual


class MixtralForCausalLM(torch.nn.Module):
    def __init__(self, cfg, device="cpu", dtype=torch.bfloat16,
                 tp: TPContext | None = None, seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.tp = tp or TPContext()
        torch.manual_seed(seed)
        self.embed = init_weight((cfg.vocab_size, cfg.hidden_size), device, dtype)
        self.layers = torch.nn.ModuleList([
            MixtralDecoderLayer(cfg, i, self.tp, device, dtype)
            for i in range(cfg.num_layers)
        ])
        self.final_norm_w = init_const((cfg.hidden_size,), device, dtype, 1.0)
        self.lm_head = init_weight((cfg.vocab_size, cfg.hidden_size), device, dtype)
        from dynamo_amd.ops import torch_ref
        self.cos_sin = torch_ref.make_cos_sin_cache(
            cfg.max_position, cfg.head_dim, cfg.rope_theta, device=device)

    def forward(self, input_ids, kv_pool, meta: AttnMetadata):
        x = F.embedding(input_ids.long(), self.embed)
        if meta.inputs_embeds is not None:
            x = x.index_copy(0, meta.embeds_rows,
                             meta.inputs_embeds.to(x.dtype))
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer.forward(x, residual, self.cos_sin,
                                        kv_pool.kcache(i), kv_pool.vcache(i),
                                        meta)
        x = ops.fused_add_rmsnorm(x, residual, self.final_norm_w, self.cfg.rms_eps)
        return x

    def compute_logits(self, hidden):
        return linear(hidden, self.lm_head).float()
