"""Model runner: turns a SchedulerOutput into one forward pass + sampling.

Builds the flattened token batch (decode tokens first, then prefill chunks),
the attention metadata (page tables, slot mappings, varlen descriptors) and
runs the model. hipGraph capture for decode-only steps is handled by
GraphRunner (engine/graphs.py) layered on top of this.
"""
from __future__ import annotations

import time
from typing import List, Optional, Tuple

import torch

from dynamo_amd import ops
from dynamo_amd.models.layers import AttnMetadata, TPContext
from .config import EngineConfig
from .kv_cache import KVCachePool
from .scheduler import SchedulerOutput, ScheduledSeq


class ModelRunner:
    def __init__(self, cfg: EngineConfig, tp: Optional[TPContext] = None,
                 seed: int = 0, weight_pool=None):
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        self.dtype = cfg.torch_dtype
        self.tp = tp or TPContext(cfg.tp_size, cfg.tp_rank)
        m = cfg.model
        from dynamo_amd.models.registry import build_model
        if weight_pool is not None:
            from dynamo_amd.gms import weight_allocator
            with weight_allocator(weight_pool):
                self.model = build_model(m, self.device, self.dtype, self.tp,
                                         seed)
        else:
            self.model = build_model(m, self.device, self.dtype, self.tp, seed)
        if getattr(m, "weights_path", ""):
            from dynamo_amd.models.loader import load_weights
            n = load_weights(self.model, m.weights_path)
            import logging
            logging.getLogger("dynamo_amd.engine").info(
                "loaded %d checkpoint tensors from %s", n, m.weights_path)
        self.hkv_local = max(1, m.num_kv_heads // self.tp.size)
        self.hq_local = m.num_q_heads // self.tp.size

        num_pages = cfg.kv_pool_pages or self._auto_pages()
        # resolve the V-page layout against the LOCAL (TP-sharded) head
        # counts - the swapped decode kernel needs 2 <= G_local <= 16
        g_local = (self.hq_local // self.hkv_local
                   if self.hq_local % self.hkv_local == 0 else 0)
        self.v_transposed = cfg.v_transposed and 2 <= g_local <= 16
        self.kv_pool = KVCachePool(m.num_layers, num_pages, self.hkv_local,
                                   cfg.page_size, m.head_dim, cfg.device,
                                   cfg.kv_torch_dtype,
                                   shm_export=cfg.cpu_shm_pool,
                                   v_transposed=self.v_transposed)
        self.num_pages = num_pages
        self.max_pages_per_seq = (cfg.max_model_len + cfg.page_size - 1) // cfg.page_size
        self.decode_scratch = None
        if self.device.type == "cuda":
            self.decode_scratch = ops.DecodeScratch(
                cfg.max_num_seqs, self.hq_local, m.head_dim,
                cfg.max_model_len, self.device)

    def _auto_pages(self) -> int:
        m = self.cfg.model
        page_bytes = (m.num_layers * 2 * self.hkv_local * self.cfg.page_size
                      * m.head_dim * 2)
        if self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            budget = int(free * self.cfg.gpu_mem_fraction) - (2 << 30)
        else:
            budget = 1 << 30  # 1 GiB for CPU tests
        return max(16, budget // page_bytes)

    # ------------------------------------------------------------------
    def prepare(self, sched: SchedulerOutput) -> Tuple[torch.Tensor, AttnMetadata]:
        cfg = self.cfg
        ps = cfg.page_size
        dev = self.device
        tokens: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        logits_rows: List[int] = []

        # ---- decode part ----
        nd = len(sched.decodes)
        dec_tables, dec_ctx = [], []
        for i, ss in enumerate(sched.decodes):
            r = ss.req
            pos = r.num_computed
            tokens.append(r.all_tokens[pos])
            positions.append(pos)
            pages = r.kv.pages
            slots.append(pages[pos // ps] * ps + pos % ps)
            dec_tables.append(pages)
            dec_ctx.append(pos + 1)
            logits_rows.append(i)

        # ---- prefill part ----
        pf_tables, q_start, q_len, ctx_len = [], [], [], []
        embeds_rows, embeds_parts = [], []
        qpos = 0
        for ss in sched.prefills:
            r = ss.req
            nc, n = r.num_computed, ss.n_new
            seq_tokens = r.all_tokens
            for j in range(nc, nc + n):
                tokens.append(seq_tokens[j])
                positions.append(j)
                slots.append(r.kv.pages[j // ps] * ps + j % ps)
            if r.prompt_embeds is not None:
                # rows of this chunk that fall inside the prompt get their
                # embeddings from the request, not the token table
                pe = min(nc + n, len(r.prompt_tokens)) - nc
                if pe > 0:
                    base = nd + qpos
                    embeds_rows.extend(range(base, base + pe))
                    embeds_parts.append(r.prompt_embeds[nc:nc + pe])
            if r.embed_spans:
                # multimodal: sparse spans (image placeholder rows) that
                # overlap this prefill chunk take encoder embeddings
                for off, emb in r.embed_spans:
                    lo = max(nc, off)
                    hi = min(nc + n, off + emb.shape[0])
                    if lo < hi:
                        base = nd + qpos + (lo - nc)
                        embeds_rows.extend(range(base, base + hi - lo))
                        embeds_parts.append(emb[lo - off:hi - off])
            pf_tables.append(r.kv.pages)
            q_start.append(qpos)
            q_len.append(n)
            ctx_len.append(nc + n)
            if ss.sample and not r.sampling.embed:
                logits_rows.append(nd + qpos + n - 1)
            qpos += n

        def table_tensor(tables):
            if not tables:
                return None
            maxp = max(len(t) for t in tables)
            out = torch.zeros(len(tables), max(1, maxp), dtype=torch.int32)
            for i, t in enumerate(tables):
                out[i, :len(t)] = torch.tensor(t, dtype=torch.int32)
            return out.to(dev, non_blocking=True)

        meta = AttnMetadata(
            slot_mapping=torch.tensor(slots, dtype=torch.int64).to(dev, non_blocking=True),
            positions=torch.tensor(positions, dtype=torch.int32).to(dev, non_blocking=True),
            num_decode=nd,
            decode_page_table=table_tensor(dec_tables),
            decode_ctx_lens=(torch.tensor(dec_ctx, dtype=torch.int32).to(dev, non_blocking=True)
                             if dec_ctx else None),
            decode_scratch=self.decode_scratch,
            num_prefill_tokens=qpos,
            prefill_page_table=table_tensor(pf_tables),
            seq_q_start=(torch.tensor(q_start, dtype=torch.int32).to(dev, non_blocking=True)
                         if q_start else None),
            seq_q_len=(torch.tensor(q_len, dtype=torch.int32).to(dev, non_blocking=True)
                       if q_len else None),
            seq_ctx_len=(torch.tensor(ctx_len, dtype=torch.int32).to(dev, non_blocking=True)
                         if ctx_len else None),
            logits_rows=torch.tensor(logits_rows, dtype=torch.int64).to(dev, non_blocking=True),
            v_transposed=self.v_transposed,
        )
        if embeds_rows:
            meta.embeds_rows = torch.tensor(embeds_rows, dtype=torch.int64,
                                            device=dev)
            meta.inputs_embeds = torch.cat(embeds_parts, 0).to(dev)
        if qpos and dev.type == "cuda":
            rows = ops.prefill_tile_rows(self.hq_local, self.hkv_local)
            meta.prefill_tiles = ops.build_prefill_tiles(q_len, dev, rows)
        input_ids = torch.tensor(tokens, dtype=torch.int32).to(dev, non_blocking=True)
        return input_ids, meta

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def execute(self, sched: SchedulerOutput, step_seed: int = 0):
        """Returns (sampled_tokens int32 [n_sample] on device, sample_reqs)."""
        input_ids, meta = self.prepare(sched)
        hidden = self.model.forward(input_ids, self.kv_pool, meta)
        # embedding requests: accumulate fp32 hidden-state sums per chunk
        # (mean pool finalized by the engine when the prefill completes)
        qpos = 0
        for ss in sched.prefills:
            r = ss.req
            if r.sampling.embed:
                seg = hidden[meta.num_decode + qpos:
                             meta.num_decode + qpos + ss.n_new].float().sum(0)
                r._embed_sum = (seg if r._embed_sum is None
                                else r._embed_sum + seg)
            qpos += ss.n_new
        sample_seqs = [s for s in sched.decodes if s.sample] + \
                      [s for s in sched.prefills
                       if s.sample and not s.req.sampling.embed]
        if not sample_seqs:
            return torch.empty(0, dtype=torch.int32), []
        rows = hidden[meta.logits_rows]
        logits = self.model.compute_logits(rows)
        from .sampling import compute_logprobs, sample_tokens
        reqs = [s.req for s in sample_seqs]
        sampled = sample_tokens(logits, reqs, step_seed)
        for r, lp in zip(reqs, compute_logprobs(logits, sampled, reqs)):
            r._logprobs = lp
        return sampled, reqs
