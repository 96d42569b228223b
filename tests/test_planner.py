"""Planner tests: predictors, load scaling with hysteresis, SLA
throughput sizing, and the metrics-polling service over mock workers."""
import asyncio
import time

import pytest

from dynamo_amd.planner import (LoadPlanner, MovingAveragePredictor,
                                PerfModel, PlannerService, PoolObservation,
                                PoolPolicy, SLATargets, ThroughputPlanner,
                                TrendPredictor, VirtualConnector)


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_predictors():
    m = MovingAveragePredictor(window=4)
    for v in [1, 2, 3, 4, 5]:
        m.observe(v)
    assert m.predict() == pytest.approx(3.5)
    t = TrendPredictor(window=4)
    for v in [1, 2, 3, 4]:
        t.observe(v)
    assert t.predict() > 3.9  # rising trend extrapolates up


def test_load_planner_scales_up_and_down():
    async def main():
        conn = VirtualConnector({"backend": 2})
        p = PoolPolicy("backend", min_replicas=1, max_replicas=4,
                       cooldown_s=0.0, predictor="constant")
        pl = LoadPlanner([p], conn)
        # hot: kv usage above threshold
        await pl.observe_and_plan(
            {"backend": PoolObservation(kv_usage=0.95, replicas=2)})
        assert conn.current("backend") == 3
        # queue pressure
        await pl.observe_and_plan(
            {"backend": PoolObservation(kv_usage=0.5, num_waiting=20,
                                        replicas=3)})
        assert conn.current("backend") == 4
        # capped at max
        await pl.observe_and_plan(
            {"backend": PoolObservation(kv_usage=0.99, replicas=4)})
        assert conn.current("backend") == 4
        # cool: scale down
        for _ in range(8):
            await pl.observe_and_plan(
                {"backend": PoolObservation(kv_usage=0.05, replicas=4)})
        assert conn.current("backend") < 4
    run(main())


def test_load_planner_cooldown():
    async def main():
        conn = VirtualConnector({"backend": 1})
        p = PoolPolicy("backend", cooldown_s=60.0, predictor="constant")
        pl = LoadPlanner([p], conn)
        await pl.observe_and_plan(
            {"backend": PoolObservation(kv_usage=0.95, replicas=1)})
        assert conn.current("backend") == 2
        await pl.observe_and_plan(
            {"backend": PoolObservation(kv_usage=0.99, replicas=2)})
        assert conn.current("backend") == 2  # cooldown holds
    run(main())


def test_throughput_planner_sla_sizing():
    async def main():
        conn = VirtualConnector()
        tp = ThroughputPlanner(
            SLATargets(ttft_s=2.0, itl_ms=25.0, isl=8192, osl=1024),
            PerfModel(prefill_tokens_per_s=100_000,
                      decode_tokens_per_s_at_itl=640, max_conc_at_itl=16),
            conn, predictor="constant")
        targets = await tp.observe_and_plan(req_per_s=2.0)
        # prefill: 2 * 8192 / 100k -> 1; decode inflight: 2*1024*0.025=51.2
        # -> 51.2/16 -> 4
        assert targets["prefill"] == 1
        assert targets["backend"] == 4
        targets = await tp.observe_and_plan(req_per_s=40.0)
        assert targets["prefill"] >= 4
        assert targets["backend"] >= 32
    run(main())


def test_planner_service_with_mock_workers():
    async def main():
        from dynamo_amd.engine.config import ModelConfig
        from dynamo_amd.mocker import make_mock_engine
        from dynamo_amd.runtime import DistributedRuntime, MemoryDiscovery
        from dynamo_amd.workers import WorkerService
        from dynamo_amd.engine.scheduler import SamplingParams

        shared = MemoryDiscovery()
        rt = DistributedRuntime(shared)
        # decode_step_ms keeps the pool saturated long enough for the
        # planner's 0.1 s polling to observe it (a healthy engine on a CPU
        # mock otherwise finishes the burst in milliseconds)
        eng = make_mock_engine(model=ModelConfig(name="m", vocab_size=512),
                               num_pages=32, page_size=16,
                               decode_step_ms=50.0)
        ws = WorkerService(eng, rt)
        await ws.start()
        # saturate the tiny KV pool
        for i in range(6):
            eng.add_request(f"r{i}", list(range(60)),
                            SamplingParams(max_tokens=64, ignore_eos=True))
        ws._work.set()
        conn = VirtualConnector({"backend": 1})
        planner = LoadPlanner([PoolPolicy("backend", cooldown_s=0.0,
                                          predictor="constant")], conn)
        prt = DistributedRuntime(shared)
        svc = PlannerService(prt, "dynamo", planner, interval=0.1)
        await svc.start()
        for _ in range(50):
            if conn.current("backend") > 1:
                break
            await asyncio.sleep(0.1)
        assert conn.current("backend") > 1, "planner never scaled up"
        await svc.stop()
        await ws.stop()
        await rt.shutdown(drain=False)
        await prt.shutdown(drain=False)
    run(main())
