"""hipGraph-captured decode steps.

The decode inner loop (embed -> N layers -> norm -> lm_head) is captured
once per batch-size bucket into a hipGraph (torch.cuda.CUDAGraph == hipGraph
on ROCm) operating on persistent device buffers; steady-state decode then
costs one graph replay + one sampling kernel + one D2H token fetch, instead
of ~7xL kernel launches issued from Python.

Buffer advance (positions+1, ctx+1, slot recompute from the page table) runs
as a handful of small torch ops before each replay; page-table rows are
patched from the host only when a sequence crosses a page boundary (every
page_size steps) or the batch composition changes.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from dynamo_amd.models.layers import AttnMetadata
from .scheduler import Request

BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256]


class GraphRunner:
    def __init__(self, runner, max_batch: int, use_graphs: bool = True):
        self.runner = runner
        self.use_graphs = use_graphs
        cfg = runner.cfg
        self.cfg = cfg
        self.device = runner.device
        self.max_batch = max_batch
        self.ps = cfg.page_size
        self.max_pages = runner.max_pages_per_seq
        dev = self.device
        B = max_batch
        self.input_ids = torch.zeros(B, dtype=torch.int32, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.ctx_lens = torch.zeros(B, dtype=torch.int32, device=dev)
        self.slot_mapping = torch.zeros(B, dtype=torch.int64, device=dev)
        self.page_table = torch.zeros(B, self.max_pages, dtype=torch.int32,
                                      device=dev)
        V = cfg.model.vocab_size
        self.logits = torch.empty(B, V, dtype=torch.float32, device=dev)
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._graph_pool = None
        # current batch composition
        self.reqs: List[Request] = []
        self.dirty = True
        self._first_after_rebuild = True

    # ------------------------------------------------------------------
    def _meta(self, bc: int) -> AttnMetadata:
        return AttnMetadata(
            slot_mapping=self.slot_mapping[:bc],
            positions=self.positions[:bc],
            num_decode=bc,
            decode_page_table=self.page_table[:bc],
            decode_ctx_lens=self.ctx_lens[:bc],
            decode_scratch=self.runner.decode_scratch,
            num_prefill_tokens=0,
            v_transposed=getattr(self.runner, "v_transposed", False),
        )

    @torch.inference_mode()
    def _capture(self, bc: int):
        model = self.runner.model
        kv_pool = self.runner.kv_pool
        meta = self._meta(bc)
        torch.cuda.synchronize()
        # warm up the exact op sequence on a side stream first
        s = torch.cuda.Stream()
        with torch.cuda.stream(s):
            h = model.forward(self.input_ids[:bc], kv_pool, meta)
            self.logits[:bc] = model.compute_logits(h)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._graph_pool):
            h = model.forward(self.input_ids[:bc], kv_pool, meta)
            self.logits[:bc] = model.compute_logits(h)
        if self._graph_pool is None:
            self._graph_pool = g.pool()
        self.graphs[bc] = g

    def _bucket(self, n: int) -> int:
        for b in BUCKETS:
            if b >= n and b <= self.max_batch:
                return b
        return self.max_batch

    # ------------------------------------------------------------------
    def rebuild(self, reqs: List[Request]):
        """Full buffer rebuild after batch composition change."""
        self.reqs = list(reqs)
        n = len(reqs)
        assert n <= self.max_batch
        ids, pos, slots = [], [], []
        pt = torch.zeros(n, self.max_pages, dtype=torch.int32)
        for i, r in enumerate(reqs):
            p = r.num_computed
            ids.append(r.all_tokens[p])
            pos.append(p)
            slots.append(r.kv.pages[p // self.ps] * self.ps + p % self.ps)
            pt[i, :len(r.kv.pages)] = torch.tensor(r.kv.pages, dtype=torch.int32)
        self.input_ids[:n].copy_(torch.tensor(ids, dtype=torch.int32))
        self.positions[:n].copy_(torch.tensor(pos, dtype=torch.int32))
        self.ctx_lens[:n].copy_(torch.tensor([p + 1 for p in pos],
                                             dtype=torch.int32))
        self.slot_mapping[:n].copy_(torch.tensor(slots, dtype=torch.int64))
        self.page_table[:n].copy_(pt)
        # pad rows: ctx 0 disables attention; slot -1 skips kv append
        if n < self.max_batch:
            self.ctx_lens[n:].fill_(0)
            self.slot_mapping[n:].fill_(-1)
        self.dirty = False
        self._first_after_rebuild = True

    def patch_new_page(self, i: int, page_idx: int, page_id: int):
        """A sequence allocated a new page (host-side) — patch one cell."""
        self.page_table[i, page_idx] = page_id

    # ------------------------------------------------------------------
    @torch.inference_mode()
    def step(self, sampled_prev: Optional[torch.Tensor]) -> torch.Tensor:
        """Run one captured decode step for the current batch.

        sampled_prev: device int32 [n] tokens sampled by the previous step
        (None right after a rebuild — buffers already hold current state).
        Returns the logits rows' sampled tokens (device tensor) for this
        step via engine-side sampling.
        """
        n = len(self.reqs)
        bc = self._bucket(n)
        if not self._first_after_rebuild:
            assert sampled_prev is not None
            self.input_ids[:n].copy_(sampled_prev[:n])
            self.positions[:n] += 1
            self.ctx_lens[:n] += 1
            pos = self.positions[:n].long()
            page_idx = (pos // self.ps).unsqueeze(1)
            pages = self.page_table[:n].gather(1, page_idx).squeeze(1)
            self.slot_mapping[:n] = pages.to(torch.int64) * self.ps + pos % self.ps
        self._first_after_rebuild = False
        if not self.use_graphs:
            # persistent-buffer eager path (TP ranks: RCCL-in-graph capture
            # is not exercised; MoE: host-side tile lists aren't capturable)
            meta = self._meta(bc)
            h = self.runner.model.forward(self.input_ids[:bc],
                                          self.runner.kv_pool, meta)
            self.logits[:bc] = self.runner.model.compute_logits(h)
            return self.logits[:n]
        if bc not in self.graphs:
            self._capture(bc)
        self.graphs[bc].replay()
        return self.logits[:n]
