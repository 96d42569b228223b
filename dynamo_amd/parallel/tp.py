"""Tensor-parallel engine coordination (one process per GPU over RCCL).

Design: every TP rank runs the FULL engine (scheduler + allocator + model
shard). The scheduler is deterministic, so ranks stay in lockstep as long
as they see the same request arrivals at the same step boundaries — rank 0
broadcasts {new_requests, aborts} before each step over a gloo control
group (tiny messages; RCCL handles the heavy collectives inside the model).
Sampling is identical on all ranks (logits are identical after the final
all-reduce; greedy/seeded-Gumbel are deterministic), so no result
broadcast is needed.

This replaces the reference's engine-internal TP (vLLM's) with a native
scheme sized for xGMI: the only per-step control traffic is one small
object broadcast; weight shards are deterministic slices of the same full
tensors (models/layers.init_sharded).
"""
from __future__ import annotations

import logging
from typing import List, Optional

import torch
import torch.distributed as dist

from dynamo_amd.engine.engine import LLMEngine
from dynamo_amd.engine.scheduler import SamplingParams
from dynamo_amd.models.layers import TPContext

log = logging.getLogger("dynamo_amd.tp")


def init_tp(backend: Optional[str] = None) -> TPContext:
    """Initialize torch.distributed from torchrun env and build TPContext
    (+ a gloo side group for control messages)."""
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    ctl = dist.new_group(backend="gloo")
    tp = TPContext(world, rank, group=None)  # default group for collectives
    tp.control_group = ctl
    return tp


class TPEngineGroup:
    """Rank-0 facade over a TP engine group (LLMEngine-compatible surface
    for WorkerService / bench)."""

    def __init__(self, engine: LLMEngine, tp: TPContext):
        self.engine = engine
        self.tp = tp
        self._pending_new: List[dict] = []
        self._pending_aborts: List[str] = []
        assert tp.rank == 0, "TPEngineGroup runs on rank 0 only"
        # LLMEngine-compatible attributes
        self.cfg = engine.cfg
        self.alloc = engine.alloc
        self.scheduler = engine.scheduler
        self.requests = engine.requests
        self.runner = engine.runner

    # -- engine surface -------------------------------------------------
    def add_request(self, req_id, prompt_tokens, sampling: SamplingParams):
        self._pending_new.append({
            "request_id": req_id, "token_ids": list(prompt_tokens),
            "sampling": sampling.__dict__.copy()})
        return self.engine.add_request(req_id, prompt_tokens, sampling)

    def abort(self, req_id):
        self._pending_aborts.append(req_id)
        self.engine.abort(req_id)

    def has_work(self):
        return self.engine.has_work()

    def _broadcast(self, cmd: dict):
        dist.broadcast_object_list([cmd], src=0, group=self.tp.control_group)

    def step(self):
        cmd = {"new": self._pending_new, "aborts": self._pending_aborts}
        self._pending_new, self._pending_aborts = [], []
        self._broadcast(cmd)
        return self.engine.step()

    def shutdown(self):
        self._broadcast({"shutdown": True})

    def drain_kv_events(self):
        return self.engine.drain_kv_events()

    def clear_kv(self):
        # followers clear on the next step command
        self._pending_aborts.append("__clear_kv__")
        self.engine.clear_kv()

    def release_held(self, req_id):
        self.engine.release_held(req_id)

    @property
    def last_metrics(self):
        return self.engine.last_metrics

    @property
    def step_count(self):
        return self.engine.step_count

    @property
    def _held(self):
        return self.engine._held


def follower_loop(engine: LLMEngine, tp: TPContext):
    """Ranks 1..N-1: apply broadcast commands in lockstep with rank 0."""
    sp_fields = set(SamplingParams().__dict__)
    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0, group=tp.control_group)
        cmd = box[0]
        if cmd.get("shutdown"):
            return
        for r in cmd.get("new", []):
            sp = SamplingParams(**{k: v for k, v in r["sampling"].items()
                                   if k in sp_fields})
            engine.add_request(r["request_id"], r["token_ids"], sp)
        for rid in cmd.get("aborts", []):
            if rid == "__clear_kv__":
                engine.clear_kv()
            else:
                engine.abort(rid)
        engine.step()
