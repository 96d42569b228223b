"""TP workers through the FRAMEWORK serving path (CPU/gloo, no GPU).

VERDICT r1 item #1: BASELINE configs #3/#4 are multi-GPU and must run
through frontend -> router -> worker, not just bench.py's private
protocol. These tests launch REAL worker CLIs (torchrun for TP>1, gloo
backend), real file discovery, real TCP request plane, and drive them
through ModelManager.generate_tokens — then check TP2 == TP1 and
disagg(TP2 prefill + TP2 decode) == aggregated TP1, token for token
(reference parity: tests/kvbm_integration/test_determinism_disagg.py,
recipes/llama-3-70b/vllm/disagg-single-node/deploy.yaml TP pools).
"""
from __future__ import annotations

import asyncio
import os
import socket
import sys
import tempfile

import pytest

from proc_utils import ManagedProcess

MODEL = "tiny-llama"
PROMPT_TOKENS = list(range(40, 90))


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def torchrun_worker(nproc: int, disc: str, **kw):
    args = [sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={nproc}",
            "--master-addr=127.0.0.1", f"--master-port={free_port()}",
            "--no-python", sys.executable,
            "-m", "dynamo_amd.workers",
            "--model", MODEL, "--device", "cpu", "--dtype", "float32",
            "--discovery", disc, "--tp-size", str(nproc),
            "--page-size", "16", "--kv-pool-pages", "128",
            "--max-num-seqs", "4", "--max-batched-tokens", "256",
            "--max-model-len", "512"]
    for k, v in kw.items():
        flag = "--" + k.replace("_", "-")
        if v is True:
            args.append(flag)
        elif v is not None and v is not False:
            args += [flag, str(v)]
    return ManagedProcess(args, ready_marker="WORKER_READY", timeout=180)


def single_worker(disc: str, **kw):
    args = [sys.executable, "-m", "dynamo_amd.workers",
            "--model", MODEL, "--device", "cpu", "--dtype", "float32",
            "--discovery", disc,
            "--page-size", "16", "--kv-pool-pages", "128",
            "--max-num-seqs", "4", "--max-batched-tokens", "256",
            "--max-model-len", "512"]
    for k, v in kw.items():
        flag = "--" + k.replace("_", "-")
        if v is True:
            args.append(flag)
        elif v is not None and v is not False:
            args += [flag, str(v)]
    return ManagedProcess(args, ready_marker="WORKER_READY", timeout=120)


async def _generate_via_stack(disc: str, n_requests: int = 2,
                              max_tokens: int = 6):
    """Frontend pipeline (ModelManager) against whatever workers are
    registered in `disc`; returns per-request token lists."""
    os.environ["DYN_BYPASS_TOKEN_THRESHOLD"] = "4"  # tiny prompts: force
    from dynamo_amd.frontend.service import ModelManager  # the disagg path
    from dynamo_amd.runtime import DistributedRuntime
    rt = DistributedRuntime(disc)
    mgr = ModelManager(rt)
    await mgr.start(watch_interval=0.2)
    try:
        deadline = asyncio.get_event_loop().time() + 60
        while MODEL not in mgr.models:
            if asyncio.get_event_loop().time() > deadline:
                raise TimeoutError("model never appeared in discovery")
            await asyncio.sleep(0.2)
            await mgr.refresh()
        entry = mgr.get(MODEL)
        outs = []
        for i in range(n_requests):
            toks = []
            async for chunk in mgr.generate_tokens(
                    entry, [t + i for t in PROMPT_TOKENS],
                    sampling={"temperature": 0.0},
                    stop={"max_tokens": max_tokens, "ignore_eos": True}):
                toks.extend(chunk.get("token_ids", []))
            outs.append(toks)
        return outs
    finally:
        await mgr.stop()
        await rt.shutdown(drain=False)


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def _baseline_tp1(max_tokens=6, n_requests=2):
    """In-process TP1 engine reference for the same model/seed."""
    from dynamo_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from dynamo_amd.engine.config import PRESETS
    cfg = EngineConfig(model=PRESETS[MODEL], device="cpu", dtype="float32",
                       max_num_seqs=4, max_batched_tokens=256,
                       max_model_len=512, kv_pool_pages=128, page_size=16)
    outs = []
    for i in range(n_requests):
        eng = LLMEngine(cfg, seed=0)
        eng.add_request("r", [t + i for t in PROMPT_TOKENS],
                        SamplingParams(max_tokens=max_tokens,
                                       ignore_eos=True))
        toks = []
        while eng.has_work():
            for so in eng.step():
                toks.append(so.new_token)
        outs.append(toks)
    return outs


@pytest.mark.timeout(420)
def test_tp2_worker_through_framework():
    """config #4 shape at CPU scale: a TP2 worker launched by torchrun
    serves the request plane; output == TP1 engine."""
    with tempfile.TemporaryDirectory() as d:
        disc = f"file:{d}/disc"
        w = torchrun_worker(2, disc)
        w.start()
        try:
            outs = run(_generate_via_stack(disc))
        finally:
            w.stop()
        assert outs == _baseline_tp1(), f"TP2-serve {outs}"


@pytest.mark.timeout(600)
def test_disagg_tp2_pools_through_framework():
    """config #4 shape: TP2 prefill pool + TP2 decode pool; KV handoff
    rank-to-rank via shared pool mappings; output == aggregated TP1."""
    with tempfile.TemporaryDirectory() as d:
        disc = f"file:{d}/disc"
        wp = torchrun_worker(2, disc, worker_type="prefill")
        wd = torchrun_worker(2, disc, worker_type="decode",
                             component="backend")
        wp.start()
        wd.start()
        try:
            outs = run(_generate_via_stack(disc))
        finally:
            wp.stop()
            wd.stop()
        assert outs == _baseline_tp1(), f"disagg-TP2 {outs}"


@pytest.mark.timeout(420)
def test_disagg_tp1_processes_through_framework():
    """config #3 shape: separate prefill + decode worker PROCESSES (TP1),
    KV pulled cross-process through the shared pool mapping (the CPU
    stand-in for hipIpc); output == aggregated."""
    with tempfile.TemporaryDirectory() as d:
        disc = f"file:{d}/disc"
        wp = single_worker(disc, worker_type="prefill")
        wd = single_worker(disc, worker_type="decode", component="backend")
        wp.start()
        wd.start()
        try:
            outs = run(_generate_via_stack(disc))
        finally:
            wp.stop()
            wd.stop()
        assert outs == _baseline_tp1(), f"disagg-1p1d {outs}"


@pytest.mark.timeout(420)
def test_moe_ep_tp2_worker_through_framework():
    """config #5 shape: a Mixtral TP2 worker (EP auto-on: experts sharded,
    all-to-all-free partial-sum design) served through the framework;
    output == single-rank engine."""
    global MODEL
    saved = MODEL
    MODEL = "tiny-mixtral"
    try:
        with tempfile.TemporaryDirectory() as d:
            disc = f"file:{d}/disc"
            w = torchrun_worker(2, disc)
            w.start()
            try:
                outs = run(_generate_via_stack(disc))
            finally:
                w.stop()
            assert outs == _baseline_tp1(), f"moe-ep-serve {outs}"
    finally:
        MODEL = saved
