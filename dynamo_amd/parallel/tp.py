"""Tensor-parallel engine coordination (one process per GPU over RCCL).

Design: every TP rank runs the FULL engine (scheduler + allocator + model
shard). The scheduler is deterministic, so ranks stay in lockstep as long
as they see the same command sequence — rank 0 broadcasts a command dict
over a gloo control group before each step (tiny messages; RCCL handles
the heavy collectives inside the model). Sampling is identical on all
ranks (logits are identical after the final all-reduce; greedy and the
counter-based Gumbel streams are deterministic), so no result broadcast is
needed. Because the page allocators run in lockstep too, PAGE IDS AGREE
ACROSS RANKS — disagg KV handoff between equal-degree TP groups is a
rank-to-rank pull with the same page lists (the reference instead
broadcasts onboarded blocks inside the TP group over NCCL,
block_manager/distributed/transfer.rs:473-525).

Command protocol (applied in this order on every rank):
  attach   — disagg decode handoff: add request + pull KV shard from the
             peer prefill rank (rank_meta selects this rank's pool)
  new      — queued add_request batch (carries hold_kv for prefill role)
  aborts   — abort request ids
  release  — release_held ids (disagg prefill side, after decode pulled)
  clear_kv — allocator reset (clear_kv_blocks endpoint)
  lora     — load/unload lora (graph invalidation included)
  weight_delta — RL weight push: same seed on all ranks gives identical
             noise on replicated tensors and a valid independent delta on
             sharded ones (module iteration order is rank-invariant)
  step     — run engine.step() after applying the above

This replaces the reference's engine-internal TP (vLLM's) with a native
scheme sized for xGMI: the only per-step control traffic is one small
object broadcast.
"""
from __future__ import annotations

import logging
from typing import List, Optional

import torch
import torch.distributed as dist

from dynamo_amd.engine.engine import LLMEngine
from dynamo_amd.engine.kv_cache import SequenceKV
from dynamo_amd.engine.scheduler import SamplingParams
from dynamo_amd.models.layers import TPContext

log = logging.getLogger("dynamo_amd.tp")

_SP_FIELDS = set(SamplingParams().__dict__)


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


def _make_sp(d: dict) -> SamplingParams:
    return SamplingParams(**{k: v for k, v in d.items() if k in _SP_FIELDS})


def apply_command(engine: LLMEngine, cmd: dict, tp_rank: int = 0):
    """Apply one lockstep command batch to this rank's engine. Returns the
    step outputs (or None if the command did not step)."""
    for spec in cmd.get("attach", []):
        attach_remote(engine, spec, tp_rank)
    for r in cmd.get("new", []):
        req = engine.add_request(r["request_id"], r["token_ids"],
                                 _make_sp(r["sampling"]))
        if r.get("hold_kv"):
            req.hold_kv = True
        if r.get("arrival") is not None:
            # rank0's arrival stamp: queue-policy tie-breaks must order
            # identically on every rank
            req.arrival = r["arrival"]
        if r.get("embed_spans"):
            req.embed_spans = r["embed_spans"]
    for rid in cmd.get("aborts", []):
        engine.abort(rid)
    for rid in cmd.get("release", []):
        engine.release_held(rid)
    if cmd.get("clear_kv"):
        engine.clear_kv()
    lora = cmd.get("lora")
    if lora:
        if lora["op"] == "load":
            engine.load_lora(lora["name"], path=lora.get("path"),
                             rank=lora.get("rank", 8),
                             alpha=lora.get("alpha", 16.0),
                             seed=lora.get("seed", 0))
        else:
            engine.unload_lora(lora["name"])
    wd = cmd.get("weight_delta")
    if wd:
        engine.apply_weight_delta(wd["seed"], wd["scale"])
    if cmd.get("step"):
        return engine.step()
    return None


def attach_remote(engine: LLMEngine, spec: dict, tp_rank: int = 0):
    """Disagg decode-side KV attach executed identically on every rank:
    add the request, pull this rank's KV shard from the peer prefill
    rank's pool, mark the prompt computed."""
    from dynamo_amd.disagg.transfer import KvPuller, rank_meta
    sp = _make_sp(spec["sampling"])
    req = engine.add_request(spec["request_id"], spec["token_ids"], sp)
    num_tokens = int(spec["num_tokens"])
    kv = SequenceKV(engine.alloc, engine.cfg.block_salt)
    kv.ensure_capacity(num_tokens)
    pool = engine.runner.kv_pool
    if pool is not None and spec.get("src_meta") is not None:
        meta = rank_meta(spec["src_meta"], tp_rank)
        if engine._puller is None:
            engine._puller = KvPuller(pool)
        ps = engine.cfg.page_size
        npages = (num_tokens + ps - 1) // ps
        src_pages = [int(p) for p in spec["page_ids"]][:npages]
        engine._puller.pull(meta, src_pages, kv.pages[:npages])
    req.kv = kv
    req.num_computed = num_tokens
    if spec.get("arrival") is not None:
        req.arrival = spec["arrival"]
    if spec.get("first_token") is not None:
        req.output_tokens.append(int(spec["first_token"]))
    return req


class TPEngineGroup:
    """Rank-0 facade over a TP engine group (LLMEngine-compatible surface
    for WorkerService / bench)."""

    def __init__(self, engine: LLMEngine, tp: TPContext):
        self.engine = engine
        self.tp = tp
        self._pending_new: List[object] = []   # Request refs (hold_kv read
        self._pending_aborts: List[str] = []   # at broadcast time)
        assert tp.rank == 0, "TPEngineGroup runs on rank 0 only"
        # LLMEngine-compatible attributes
        self.cfg = engine.cfg
        self.alloc = engine.alloc
        self.scheduler = engine.scheduler
        self.requests = engine.requests
        self.runner = engine.runner

    # -- engine surface -------------------------------------------------
    def add_request(self, req_id, prompt_tokens, sampling: SamplingParams,
                    prompt_embeds=None):
        assert prompt_embeds is None, "prompt_embeds unsupported at TP>1"
        req = self.engine.add_request(req_id, prompt_tokens, sampling)
        self._pending_new.append(req)
        return req

    def abort(self, req_id):
        self._pending_aborts.append(req_id)
        self.engine.abort(req_id)

    def has_work(self):
        return self.engine.has_work()

    def _broadcast(self, cmd: dict):
        dist.broadcast_object_list([cmd], src=0, group=self.tp.control_group)

    def _pending_cmd(self, step: bool) -> dict:
        cmd = {"new": [{"request_id": r.req_id,
                        "token_ids": list(r.prompt_tokens),
                        "sampling": r.sampling.__dict__.copy(),
                        "hold_kv": bool(getattr(r, "hold_kv", False)),
                        # (offset, tensor) pairs pickle through the gloo
                        # object broadcast; identical on every rank
                        "embed_spans": r.embed_spans,
                        "arrival": r.arrival}
                       for r in self._pending_new],
               "aborts": self._pending_aborts, "step": step}
        self._pending_new, self._pending_aborts = [], []
        return cmd

    def _immediate(self, cmd: dict):
        """Broadcast + apply a non-step command NOW (callers hold the
        worker's engine lock, so this cannot interleave with step()).
        Pending adds/ABORTS are flushed first: rank 0 already applied them
        (aborts free pages), so followers must apply them before this
        command or allocator page ids diverge."""
        if self._pending_new or self._pending_aborts:
            self._broadcast(self._pending_cmd(step=False))
        self._broadcast(cmd)
        return apply_command(self.engine, cmd, tp_rank=0)

    def step(self):
        self._broadcast(self._pending_cmd(step=True))
        return self.engine.step()

    def shutdown(self):
        self._broadcast({"shutdown": True})

    def drain_kv_events(self):
        return self.engine.drain_kv_events()

    def clear_kv(self):
        self._immediate({"clear_kv": True})

    def release_held(self, req_id):
        self._immediate({"release": [req_id]})

    def attach_request(self, spec: dict):
        self._immediate({"attach": [spec]})
        return self.engine.requests.get(spec["request_id"])

    def load_lora(self, name, path=None, rank=8, alpha=16.0, seed=0):
        self._immediate({"lora": {"op": "load", "name": name, "path": path,
                                  "rank": rank, "alpha": alpha,
                                  "seed": seed}})

    def unload_lora(self, name):
        self._immediate({"lora": {"op": "unload", "name": name}})

    def list_loras(self):
        return self.engine.list_loras()

    def apply_weight_delta(self, seed: int, scale: float) -> int:
        self._immediate({"weight_delta": {"seed": seed, "scale": scale}})
        return getattr(self.engine, "_last_weight_delta_count", 0)

    def _invalidate_graphs(self):
        pass  # each rank invalidates its own graphs inside apply_command

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
    while True:
        box = [None]
        dist.broadcast_object_list(box, src=0, group=tp.control_group)
        cmd = box[0]
        if cmd.get("shutdown"):
            return
        apply_command(engine, cmd, tp_rank=tp.rank)
        engine.drain_kv_events()   # rank 0 publishes; drop ours
