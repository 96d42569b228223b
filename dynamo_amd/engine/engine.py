"""LLMEngine: the native continuous-batching engine loop.

One engine per GPU (TP>1: one engine per rank, rank 0 drives scheduling and
broadcasts — see workers/). Produces per-step outputs, KV events for the
router (stored/removed block hashes) and ForwardPassMetrics for the planner.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from dynamo_amd.models.layers import TPContext
from .config import EngineConfig
from .kv_cache import KvEvent, PageAllocator
from .model_runner import ModelRunner
from .scheduler import Request, ReqState, SamplingParams, Scheduler


@dataclass
class StepOutput:
    req_id: str
    new_token: Optional[int]
    finished: bool
    finish_reason: Optional[str] = None
    num_output_tokens: int = 0
    embedding: Optional[List[float]] = None
    logprobs: Optional[dict] = None   # {"token_logprob", "top"} if requested


@dataclass
class ForwardPassMetrics:
    """Per-iteration scheduler/engine metrics for the planner (mirrors the
    reference's ForwardPassMetrics schema,
    components/src/dynamo/common/forward_pass_metrics.py:153)."""
    worker_id: str = ""
    step: int = 0
    num_running: int = 0
    num_waiting: int = 0
    kv_usage: float = 0.0
    num_tokens_step: int = 0
    prefill_tokens_step: int = 0
    decode_tokens_step: int = 0
    step_time_ms: float = 0.0
    total_kv_pages: int = 0


class LLMEngine:
    def __init__(self, cfg: EngineConfig, tp: Optional[TPContext] = None,
                 seed: int = 0, runner=None, weight_pool=None):
        self.cfg = cfg
        self.runner = (runner if runner is not None
                       else ModelRunner(cfg, tp, seed, weight_pool=weight_pool))
        self.alloc = PageAllocator(self.runner.num_pages, cfg.page_size,
                                   cfg.enable_prefix_caching)
        self.host_tier = None
        if cfg.host_cache_pages > 0 and self.runner.kv_pool is not None:
            from dynamo_amd.kvbm.host_tier import HostKVTier
            self.host_tier = HostKVTier(self.runner.kv_pool,
                                        cfg.host_cache_pages,
                                        disk_path=cfg.disk_cache_path,
                                        num_disk_pages=cfg.disk_cache_pages,
                                        object_dir=cfg.object_cache_dir,
                                        policy=cfg.host_cache_policy)
            self.alloc.host_tier = self.host_tier
        self.scheduler = Scheduler(cfg, self.alloc)
        self.requests: Dict[str, Request] = {}
        self.step_count = 0
        self.kv_events: List[KvEvent] = []
        self.last_metrics = ForwardPassMetrics()
        self._held: Dict[str, Request] = {}  # finished but KV retained (disagg)
        # hipGraph decode fast path
        self.graph_runner = None
        if cfg.enable_hip_graphs and self.runner.device.type == "cuda":
            from .graphs import GraphRunner
            # full hipGraph capture for TP=1 (MoE decode routing is fully
            # device-side, so it captures too); TP ranks use the
            # persistent-buffer eager fast path (RCCL-in-graph untested)
            import os as _os
            moe_graphs = _os.environ.get("DYNAMO_MOE_GRAPHS", "1") != "0"
            use_graphs = cfg.tp_size == 1 and (
                cfg.model.num_experts == 0 or moe_graphs)
            self.graph_runner = GraphRunner(self.runner, cfg.max_num_seqs,
                                            use_graphs=use_graphs)
        self._last_sampled = None
        self._lora = None
        self._puller = None   # disagg decode-side KvPuller (lazy)
        self._last_weight_delta_count = 0

    # -- LoRA (engine-level activation; see dynamo_amd/lora) -----------
    @property
    def lora(self):
        if self._lora is None and hasattr(self.runner, "model"):
            from dynamo_amd.lora import LoRAManager
            self._lora = LoRAManager(self.runner.model)
        return self._lora

    def _invalidate_graphs(self):
        if self.graph_runner is not None:
            self.graph_runner.graphs.clear()
            self.graph_runner.dirty = True

    def load_lora(self, name, path=None, rank=8, alpha=16.0, seed=0,
                  activate=True):
        self.lora.load(name, path=path, rank=rank, alpha=alpha, seed=seed)
        if activate:
            self.lora.activate(name)
            self._invalidate_graphs()

    def unload_lora(self, name):
        self.lora.unload(name)
        self._invalidate_graphs()

    def list_loras(self):
        return [] if self._lora is None else self._lora.list()

    # ------------------------------------------------------------------
    def add_request(self, req_id: str, prompt_tokens: List[int],
                    sampling: SamplingParams | None = None,
                    prompt_embeds=None) -> Request:
        if prompt_embeds is not None and not prompt_tokens:
            prompt_tokens = [0] * prompt_embeds.shape[0]
        req = Request(req_id, prompt_tokens, sampling or SamplingParams())
        req.prompt_embeds = prompt_embeds
        self.requests[req_id] = req
        self.scheduler.add_request(req)
        return req

    def abort(self, req_id: str):
        self.scheduler.abort(req_id)
        self.requests.pop(req_id, None)

    def has_work(self) -> bool:
        return self.scheduler.has_work()

    # ------------------------------------------------------------------
    def _check_finish(self, req: Request) -> Optional[str]:
        sp = req.sampling
        if len(req.output_tokens) >= sp.max_tokens:
            return "length"
        if req.total_len >= self.cfg.max_model_len:
            return "length"
        if (not sp.ignore_eos and sp.stop_token_ids
                and req.output_tokens
                and req.output_tokens[-1] in sp.stop_token_ids):
            return "stop"
        return None

    # ------------------------------------------------------------------
    def _fast_decode_eligible(self) -> bool:
        if self.graph_runner is None or self.scheduler.waiting:
            return False
        running = self.scheduler.running
        if not running or len(running) > self.graph_runner.max_batch:
            return False
        return all(r.is_decode and r.total_len - r.num_computed == 1
                   for r in running)

    def _fast_decode_step(self) -> List[StepOutput]:
        from dynamo_amd.observability import roctx_range
        gr = self.graph_runner
        running = self.scheduler.running
        rebuilt = gr.dirty or gr.reqs != running
        # host-side page growth (every page_size steps per seq)
        for i, r in enumerate(running):
            pos = r.num_computed
            need = pos // self.cfg.page_size + 1
            if len(r.kv.pages) < need:
                try:
                    r.kv.ensure_capacity(pos + 1)
                except MemoryError:
                    return None  # let the eager path preempt
                if not rebuilt:
                    gr.patch_new_page(i, len(r.kv.pages) - 1, r.kv.pages[-1])
        if rebuilt:
            gr.rebuild(running)
        self.step_count += 1
        t0 = time.monotonic()
        if self.host_tier is not None:
            self.host_tier.fence()
        with roctx_range("decode_step"):
            logits = gr.step(self._last_sampled)
        from .sampling import compute_logprobs, sample_tokens
        sampled = sample_tokens(logits, running, self.step_count)
        lps = compute_logprobs(logits, sampled, running)
        self._last_sampled = sampled
        toks = sampled.cpu().tolist()

        outputs: List[StepOutput] = []
        finished_any = False
        for tok, req in zip(toks, list(running)):
            req.num_computed += 1
            req.output_tokens.append(int(tok))
            if req.first_token_time is None:
                req.first_token_time = time.monotonic()
            reason = self._check_finish(req)
            finished = reason is not None
            if finished:
                finished_any = True
                if req.hold_kv:
                    req.state = ReqState.FINISHED
                    req.finish_reason = reason
                    req.finish_time = time.monotonic()
                    self.scheduler.running.remove(req)
                    self._held[req.req_id] = req
                else:
                    self._finish(req, reason)
            elif (self.cfg.kv_events or self.cfg.enable_prefix_caching) and \
                    not req.has_embeds and \
                    req.num_computed % self.cfg.page_size == 0:
                req.kv.commit_full_pages(req.all_tokens, req.num_computed)
            outputs.append(StepOutput(req.req_id, int(tok), finished, reason,
                                      len(req.output_tokens),
                                      logprobs=lps[len(outputs)]))
        if finished_any:
            gr.dirty = True
        self.kv_events.extend(self.alloc.drain_events())
        n = len(outputs)
        self.last_metrics = ForwardPassMetrics(
            step=self.step_count, num_running=self.scheduler.num_running(),
            num_waiting=0, kv_usage=self.alloc.usage, num_tokens_step=n,
            decode_tokens_step=n,
            step_time_ms=(time.monotonic() - t0) * 1000,
            total_kv_pages=self.alloc.num_pages)
        return outputs

    def step(self) -> List[StepOutput]:
        if self._fast_decode_eligible():
            out = self._fast_decode_step()
            if out is not None:
                return out
        if self.graph_runner is not None:
            self.graph_runner.dirty = True
            self._last_sampled = None
        t0 = time.monotonic()
        sched = self.scheduler.schedule()
        if sched.is_empty:
            return []
        self.step_count += 1
        if self.host_tier is not None:
            self.host_tier.fence()
        from dynamo_amd.observability import roctx_range
        with roctx_range(f"engine_step[{len(sched.decodes)}d+"
                         f"{sum(s.n_new for s in sched.prefills)}p]"):
            sampled, sample_reqs = self.runner.execute(sched, self.step_count)
        sampled = sampled.cpu().tolist() if len(sample_reqs) else []

        # advance computed counts
        for ss in sched.seqs:
            ss.req.num_computed += ss.n_new

        outputs: List[StepOutput] = []
        for tok, req in zip(sampled, sample_reqs):
            req.output_tokens.append(int(tok))
            if req.first_token_time is None:
                req.first_token_time = time.monotonic()
            reason = self._check_finish(req)
            finished = reason is not None
            if finished:
                hold = req.hold_kv if hasattr(req, "hold_kv") else False
                if hold:
                    # keep pages alive for disagg transfer
                    req.state = ReqState.FINISHED
                    req.finish_reason = reason
                    req.finish_time = time.monotonic()
                    self.scheduler.running.remove(req)
                    self._held[req.req_id] = req
                else:
                    self._finish(req, reason)
            outputs.append(StepOutput(req.req_id, int(tok), finished, reason,
                                      len(req.output_tokens),
                                      logprobs=getattr(req, "_logprobs", None)))
            req._logprobs = None

        # embedding requests: finish when the (chunked) prefill completes
        for ss in sched.prefills:
            r = ss.req
            if (r.sampling.embed and r.num_computed >= r.total_len
                    and r.state == ReqState.RUNNING):
                pooled = (r._embed_sum / float(r.total_len)).cpu().tolist()
                r.embedding = pooled
                r._embed_sum = None
                self._finish(r, "embed")
                outputs.append(StepOutput(r.req_id, None, True, "embed", 0,
                                          embedding=pooled))

        # prefix-cache hash registration + KV events
        if self.cfg.kv_events or self.cfg.enable_prefix_caching:
            for ss in sched.seqs:
                r = ss.req
                if r.kv is not None and not r.has_embeds:
                    r.kv.commit_full_pages(r.all_tokens, r.num_computed)
        self.kv_events.extend(self.alloc.drain_events())

        dt = (time.monotonic() - t0) * 1000
        self.last_metrics = ForwardPassMetrics(
            step=self.step_count,
            num_running=self.scheduler.num_running(),
            num_waiting=self.scheduler.num_waiting(),
            kv_usage=self.alloc.usage,
            num_tokens_step=sched.num_tokens,
            prefill_tokens_step=sum(s.n_new for s in sched.prefills),
            decode_tokens_step=len(sched.decodes),
            step_time_ms=dt,
            total_kv_pages=self.alloc.num_pages,
        )
        return outputs

    def _finish(self, req: Request, reason: str):
        # commit full pages before release so prefix cache retains them
        if req.kv is not None and not req.has_embeds:
            req.kv.commit_full_pages(req.all_tokens, req.num_computed)
        self.scheduler.finish(req, reason)

    def release_held(self, req_id: str):
        """Release KV of a finished-but-held request (disagg prefill side)."""
        req = self._held.pop(req_id, None)
        if req is not None and req.kv is not None:
            req.kv.commit_full_pages(req.all_tokens, req.num_computed)
            req.kv.release()
            req.kv = None

    def attach_request(self, spec: dict):
        """Disagg decode-side handoff: add a request whose prefill ran on a
        peer worker and pull its KV pages (see parallel/tp.attach_remote;
        TP groups broadcast the same spec to every rank)."""
        from dynamo_amd.parallel.tp import attach_remote
        return attach_remote(self, spec, tp_rank=self.cfg.tp_rank)

    def apply_weight_delta(self, seed: int, scale: float) -> int:
        """RL weight-update surface (reference: lib/rl/src/lib.rs:4-16):
        apply a deterministic seeded in-place delta to every weight tensor
        (modeling a policy push without checkpoint files), then invalidate
        captured graphs and flush the KV/prefix cache — cached KV computed
        under the old weights is invalid."""
        model = getattr(self.runner, "model", None)
        count = 0
        if model is not None:
            seen = set()
            with torch.no_grad():
                for mod in model.modules():
                    for name, val in sorted(vars(mod).items()):
                        if (isinstance(val, torch.Tensor)
                                and val.is_floating_point() and val.numel()
                                and name != "cos_sin"  # rope table
                                and id(val) not in seen):  # tied tensors once
                            seen.add(id(val))
                            g = torch.Generator(device="cpu")
                            g.manual_seed(seed * 1000003 + count)
                            noise = torch.randn(val.shape, generator=g,
                                                dtype=torch.float32)
                            val.add_(noise.to(val.device, val.dtype),
                                     alpha=scale)
                            count += 1
        self.clear_kv()
        self._invalidate_graphs()
        self._last_weight_delta_count = count
        return count

    def drain_kv_events(self) -> List[KvEvent]:
        if self.host_tier is not None:
            for kind, h in self.host_tier.drain_events():
                self.kv_events.append(KvEvent(kind, [h]))
        ev, self.kv_events = self.kv_events, []
        return ev

    def clear_kv(self):
        """clear_kv_blocks endpoint parity (reference: vllm/main.py
        clear_kv_blocks)."""
        self.alloc.clear()
        self.kv_events.extend(self.alloc.drain_events())
