"""GPU-free mock engine: the test backbone for router/planner/frontend E2E.

The reference ships a Rust mocker that simulates vLLM-style scheduling, KV
accounting and timing with no GPU (ai-dynamo/dynamo lib/mocker/README.md:3-10)
as the backbone of its CPU-only CI; ours reuses the REAL scheduler, page
allocator, prefix cache and KV-event machinery (dynamo_amd.engine) and
replaces only the model execution with a timing model + deterministic
token synthesis.
"""
from __future__ import annotations

import hashlib
import time
from typing import Optional

import torch

from .engine.config import EngineConfig, ModelConfig
from .engine.engine import LLMEngine
from .engine.scheduler import SchedulerOutput


def _token_for(req_id: str, pos: int, vocab: int) -> int:
    # stay in [3, min(vocab,256)): plain byte ids that decode to text and
    # can never collide with the byte tokenizer's specials (eos_id 257) —
    # a mock stream must not trigger an accidental eos stop
    hi = min(vocab, 256)
    h = hashlib.blake2b(f"{req_id}:{pos}".encode(), digest_size=8).digest()
    return 3 + int.from_bytes(h, "little") % (hi - 3)


class MockRunner:
    """ModelRunner lookalike: deterministic tokens + configurable timing."""

    def __init__(self, cfg: EngineConfig, prefill_tps: float = 0.0,
                 decode_step_ms: float = 0.0):
        self.cfg = cfg
        self.device = torch.device("cpu")
        self.num_pages = cfg.kv_pool_pages or 1024
        self.max_pages_per_seq = ((cfg.max_model_len + cfg.page_size - 1)
                                  // cfg.page_size)
        self.prefill_tps = prefill_tps
        self.decode_step_ms = decode_step_ms
        # a real (tiny) CPU pool so disagg KV transfer paths are exercised
        from .engine.kv_cache import KVCachePool
        m = cfg.model
        self.kv_pool = KVCachePool(
            min(m.num_layers, 2), self.num_pages, 1, cfg.page_size, 16, "cpu")

    def execute(self, sched: SchedulerOutput, step_seed: int = 0):
        # simulate compute time
        delay = 0.0
        if self.decode_step_ms:
            delay += self.decode_step_ms / 1000.0
        if self.prefill_tps:
            pf_tokens = sum(s.n_new for s in sched.prefills)
            delay += pf_tokens / self.prefill_tps
        if delay:
            time.sleep(delay)
        for s in sched.prefills:
            if s.req.sampling.embed:   # deterministic fake pooled hidden
                seg = torch.full((self.cfg.model.hidden_size,),
                                 float(s.n_new), dtype=torch.float32)
                s.req._embed_sum = (seg if s.req._embed_sum is None
                                    else s.req._embed_sum + seg)
        sample_seqs = [s for s in sched.decodes if s.sample] + \
                      [s for s in sched.prefills
                       if s.sample and not s.req.sampling.embed]
        toks = [_token_for(s.req.req_id, s.req.total_len,
                           self.cfg.model.vocab_size) for s in sample_seqs]
        return (torch.tensor(toks, dtype=torch.int32),
                [s.req for s in sample_seqs])


def make_mock_engine(model: ModelConfig | None = None, num_pages: int = 1024,
                     page_size: int = 16, max_num_seqs: int = 64,
                     max_batched_tokens: int = 4096, max_model_len: int = 16384,
                     prefill_tps: float = 0.0, decode_step_ms: float = 0.0,
                     worker_type: str = "aggregated") -> LLMEngine:
    cfg = EngineConfig(
        model=model or ModelConfig(name="mock", vocab_size=32000),
        device="cpu", page_size=page_size, max_num_seqs=max_num_seqs,
        max_batched_tokens=max_batched_tokens, max_model_len=max_model_len,
        kv_pool_pages=num_pages, enable_hip_graphs=False,
        worker_type=worker_type)
    return LLMEngine(cfg, runner=MockRunner(cfg, prefill_tps, decode_step_ms))
