"""Continuous-batching scheduler with chunked prefill, prefix caching and
preemption.

Semantics follow the behavior the reference orchestrates in its engines and
simulates in its GPU-free mocker (ai-dynamo/dynamo lib/mocker: vLLM-style
scheduler with KV accounting, preemption, chunked prefill) — re-implemented
natively for our engine. Token budget per step = max_batched_tokens; running
decodes are scheduled first (1 token each), then prefill chunks.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from enum import Enum
from typing import Dict, List, Optional

from .config import EngineConfig
from .kv_cache import PageAllocator, SequenceKV


@dataclass
class SamplingParams:
    max_tokens: int = 128
    temperature: float = 0.0          # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0
    stop_token_ids: List[int] = field(default_factory=list)
    ignore_eos: bool = False
    seed: int = 0
    # embedding request: prefill-only, mean-pooled hidden state instead of
    # sampled tokens (reference parity: /v1/embeddings route,
    # lib/llm/src/http service embeddings handler)
    embed: bool = False
    # top-N logprobs per generated token (0 = off); reference parity:
    # LLMEngineOutput log_probs/top_logprobs (protocols/common/llm_backend.rs)
    logprobs: int = 0


class ReqState(Enum):
    WAITING = 0
    RUNNING = 1
    FINISHED = 2


class Request:
    def __init__(self, req_id: str, prompt_tokens: List[int],
                 sampling: SamplingParams, arrival: float | None = None):
        self.req_id = req_id
        self.prompt_tokens = list(prompt_tokens)
        self.sampling = sampling
        self.output_tokens: List[int] = []
        self.num_computed = 0          # tokens with KV in cache
        self.kv: Optional[SequenceKV] = None
        self.state = ReqState.WAITING
        self.finish_reason: Optional[str] = None
        self.arrival = arrival if arrival is not None else time.monotonic()
        self.first_token_time: Optional[float] = None
        self.finish_time: Optional[float] = None
        self.num_preemptions = 0
        # disaggregation: set on decode-side requests whose prefill ran remotely
        self.prefill_result: Optional[dict] = None
        # disaggregation: prefill-side — keep KV pages alive after finish
        self.hold_kv = False
        # embedding request accumulators (mean pool over prompt tokens)
        self.embedding: Optional[List[float]] = None
        self._embed_sum = None
        # prompt_embeds input (PreprocessedRequest parity): [T, hidden]
        # tensor replacing the token-table embeddings of the prompt;
        # prefix caching is disabled for these requests (placeholder ids
        # must not produce hash hits)
        self.prompt_embeds = None
        # multimodal: sparse embedding spans [(offset, tensor), ...] —
        # image-placeholder rows of the prompt get encoder output instead
        # of token-table embeddings (reference: preprocessor.rs:2248 media
        # + encode-worker embeddings spliced into the P/D request)
        self.embed_spans = None

    @property
    def has_embeds(self) -> bool:
        return self.prompt_embeds is not None or bool(self.embed_spans)

    @property
    def all_tokens(self) -> List[int]:
        return self.prompt_tokens + self.output_tokens

    @property
    def total_len(self) -> int:
        return len(self.prompt_tokens) + len(self.output_tokens)

    @property
    def is_decode(self) -> bool:
        return self.num_computed >= len(self.prompt_tokens)

    def __repr__(self):
        return (f"Request({self.req_id}, prompt={len(self.prompt_tokens)}, "
                f"out={len(self.output_tokens)}, computed={self.num_computed}, "
                f"{self.state.name})")


@dataclass
class ScheduledSeq:
    req: Request
    n_new: int          # tokens to compute this step
    sample: bool        # whether this step produces a sampled token


@dataclass
class SchedulerOutput:
    decodes: List[ScheduledSeq] = field(default_factory=list)
    prefills: List[ScheduledSeq] = field(default_factory=list)
    preempted: List[Request] = field(default_factory=list)

    @property
    def seqs(self) -> List[ScheduledSeq]:
        return self.decodes + self.prefills

    @property
    def num_tokens(self) -> int:
        return sum(s.n_new for s in self.seqs)

    @property
    def is_empty(self) -> bool:
        return not self.seqs


class Scheduler:
    def __init__(self, cfg: EngineConfig, allocator: PageAllocator):
        self.cfg = cfg
        self.alloc = allocator
        self.waiting: List[Request] = []
        self.running: List[Request] = []

    # ------------------------------------------------------------------
    def add_request(self, req: Request):
        if req.total_len + 8 > self.cfg.max_model_len:
            req.state = ReqState.FINISHED
            req.finish_reason = "length_error"
            return
        self.waiting.append(req)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def num_waiting(self) -> int:
        return len(self.waiting)

    def num_running(self) -> int:
        return len(self.running)

    # ------------------------------------------------------------------
    def _preempt_one(self, out: SchedulerOutput, exclude: Request) -> bool:
        """Preempt the most recently admitted running request (LCFS)."""
        for i in range(len(self.running) - 1, -1, -1):
            victim = self.running[i]
            if victim is exclude:
                continue
            self.running.pop(i)
            victim.kv.release()
            victim.kv = None
            victim.num_computed = 0
            victim.num_preemptions += 1
            victim.state = ReqState.WAITING
            self.waiting.insert(0, victim)
            out.preempted.append(victim)
            return True
        return False

    def _ensure_pages(self, req: Request, upto: int, out: SchedulerOutput) -> bool:
        """Allocate pages for `upto` tokens, preempting if needed."""
        while True:
            try:
                req.kv.ensure_capacity(upto)
                return True
            except MemoryError:
                if not self._preempt_one(out, exclude=req):
                    return False

    # ------------------------------------------------------------------
    def schedule(self) -> SchedulerOutput:
        out = SchedulerOutput()
        budget = self.cfg.max_batched_tokens

        # 1) running decodes (and in-flight chunked prefills)
        for req in list(self.running):
            if budget <= 0:
                break
            if req.state is not ReqState.RUNNING:
                continue  # preempted by an earlier iteration of this loop
            remaining = req.total_len - req.num_computed
            if remaining <= 0:
                continue
            n = min(remaining, budget)
            if not self._ensure_pages(req, req.num_computed + n, out):
                # could not fit even after preemption: preempt self
                self.running.remove(req)
                req.kv.release()
                req.kv = None
                req.num_computed = 0
                req.num_preemptions += 1
                req.state = ReqState.WAITING
                self.waiting.insert(0, req)
                out.preempted.append(req)
                continue
            sample = (req.num_computed + n == req.total_len)
            item = ScheduledSeq(req, n, sample)
            (out.decodes if (req.is_decode and n == 1) else out.prefills).append(item)
            budget -= n

        # 2) admit waiting requests (queue policy decides who goes first)
        self._order_waiting()
        while (self.waiting and budget > 0
               and len(self.running) < self.cfg.max_num_seqs):
            req = self.waiting[0]
            if req.kv is None:
                req.kv = SequenceKV(self.alloc, self.cfg.block_salt)
                if (self.cfg.enable_prefix_caching and not req.prefill_result
                        and not req.has_embeds):
                    req.num_computed = req.kv.match_prefix(req.all_tokens)
            remaining = req.total_len - req.num_computed
            n = min(remaining, budget)
            if n <= 0:
                break
            if not self._ensure_pages(req, req.num_computed + n, out):
                break
            self.waiting.pop(0)
            req.state = ReqState.RUNNING
            self.running.append(req)
            sample = (req.num_computed + n == req.total_len)
            # an admitted request whose whole PROMPT is already computed
            # (disagg decode-side attach) continues as a DECODE step, not a
            # 1-token prefill chunk: the decode attention kernel is both
            # faster and bit-aligned with the aggregated path - routing it
            # through the prefill kernel produced last-ulp different
            # layer-2+ KV at the boundary position and broke
            # disagg==aggregated bit-equality at argmax knife-edges
            if req.is_decode and n == 1:
                out.decodes.append(ScheduledSeq(req, n, sample))
            else:
                out.prefills.append(ScheduledSeq(req, n, sample))
            budget -= n

        return out

    def _order_waiting(self):
        pol = getattr(self.cfg, "queue_policy", "fcfs")
        if pol == "fcfs" or len(self.waiting) < 2:
            return
        if pol == "lcfs":
            self.waiting.sort(key=lambda r: -r.arrival)
        elif pol == "wspt":
            # shortest remaining prefill first; arrival breaks ties
            self.waiting.sort(
                key=lambda r: (r.total_len - r.num_computed, r.arrival))

    # ------------------------------------------------------------------
    def finish(self, req: Request, reason: str):
        req.state = ReqState.FINISHED
        req.finish_reason = reason
        req.finish_time = time.monotonic()
        if req.kv is not None:
            req.kv.release()
            req.kv = None
        if req in self.running:
            self.running.remove(req)

    def abort(self, req_id: str) -> bool:
        for req in self.waiting:
            if req.req_id == req_id:
                self.waiting.remove(req)
                self.finish(req, "abort")
                return True
        for req in self.running:
            if req.req_id == req_id:
                self.finish(req, "abort")
                return True
        return False
