"""SLA planner: replica autoscaling from worker metrics.

Re-creates the reference's planner core (ai-dynamo/dynamo
components/src/dynamo/planner: load-based scaling core/load_scaling.py:38 +
throughput-based scaling core/throughput_scaling.py:23-206) natively:

  - LoadPlanner: scales each pool from observed ForwardPassMetrics
    (kv-cache utilization, queue depth, step saturation) with hysteresis;
  - ThroughputPlanner: SLA-driven — computes required prefill/decode
    replicas from a measured perf model (tokens/s per replica at target
    TTFT/ITL), the same shape as the reference's prefill/decode perf
    models (core/perf_model/{prefill,decode}.py);
  - load predictors: constant / moving-average / linear-trend (the
    reference ships constant/ARIMA/prophet — predictors.py);
  - connectors decide how replicas are actually added/removed: the
    VirtualConnector mirrors the reference's virtual connector
    (connectors/virtual.py) and is the test/deployment-driver interface;
    a SubprocessConnector spawns real worker processes.
"""
from __future__ import annotations

import asyncio
import logging
import math
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Callable, Deque, Dict, List, Optional

log = logging.getLogger("dynamo_amd.planner")


# -- load prediction --------------------------------------------------------
class Predictor:
    def observe(self, v: float): ...
    def predict(self) -> float: ...


class ConstantPredictor(Predictor):
    def __init__(self):
        self.last = 0.0

    def observe(self, v):
        self.last = v

    def predict(self):
        return self.last


class MovingAveragePredictor(Predictor):
    def __init__(self, window: int = 12):
        self.buf: Deque[float] = deque(maxlen=window)

    def observe(self, v):
        self.buf.append(v)

    def predict(self):
        return sum(self.buf) / len(self.buf) if self.buf else 0.0


class TrendPredictor(Predictor):
    """Linear extrapolation over the window (one horizon step ahead)."""

    def __init__(self, window: int = 12):
        self.buf: Deque[float] = deque(maxlen=window)

    def observe(self, v):
        self.buf.append(v)

    def predict(self):
        n = len(self.buf)
        if n < 2:
            return self.buf[-1] if self.buf else 0.0
        xs = range(n)
        mx = (n - 1) / 2
        my = sum(self.buf) / n
        cov = sum((x - mx) * (y - my) for x, y in zip(xs, self.buf))
        var = sum((x - mx) ** 2 for x in xs)
        slope = cov / var if var else 0.0
        return max(0.0, my + slope * ((n - 1) - mx + 1))


PREDICTORS = {"constant": ConstantPredictor, "moving_average":
              MovingAveragePredictor, "trend": TrendPredictor}


# -- connectors -------------------------------------------------------------
class Connector:
    """Applies replica-count decisions."""

    async def scale(self, component: str, target: int): ...

    def current(self, component: str) -> int: ...


class VirtualConnector(Connector):
    """Records decisions (tests / external executors read them —
    reference connectors/virtual.py parity)."""

    def __init__(self, initial: Optional[Dict[str, int]] = None):
        self.targets: Dict[str, int] = dict(initial or {})
        self.history: List[tuple] = []

    async def scale(self, component, target):
        self.targets[component] = target
        self.history.append((time.time(), component, target))

    def current(self, component):
        return self.targets.get(component, 0)


# -- planner cores ----------------------------------------------------------
@dataclass
class PoolPolicy:
    component: str
    min_replicas: int = 1
    max_replicas: int = 8
    kv_high: float = 0.85        # scale up above
    kv_low: float = 0.35         # scale down below
    queue_high: int = 4          # waiting reqs per replica
    predictor: str = "moving_average"
    cooldown_s: float = 10.0


@dataclass
class PoolObservation:
    kv_usage: float = 0.0
    num_waiting: int = 0
    num_running: int = 0
    replicas: int = 0


class LoadPlanner:
    """Reactive scaling from pool metrics with hysteresis + cooldown."""

    def __init__(self, policies: List[PoolPolicy], connector: Connector):
        self.policies = {p.component: p for p in policies}
        self.connector = connector
        self.predictors = {
            c: {"kv": PREDICTORS[p.predictor](),
                "queue": PREDICTORS[p.predictor]()}
            for c, p in self.policies.items()}
        self.last_action: Dict[str, float] = {}

    async def observe_and_plan(self, obs: Dict[str, PoolObservation]):
        for comp, p in self.policies.items():
            o = obs.get(comp)
            if o is None:
                continue
            pred = self.predictors[comp]
            pred["kv"].observe(o.kv_usage)
            pred["queue"].observe(o.num_waiting / max(1, o.replicas))
            now = time.time()
            if now - self.last_action.get(comp, 0) < p.cooldown_s:
                continue
            cur = self.connector.current(comp) or o.replicas
            kv = pred["kv"].predict()
            qd = pred["queue"].predict()
            target = cur
            if kv > p.kv_high or qd > p.queue_high:
                target = min(p.max_replicas, cur + 1)
            elif kv < p.kv_low and qd == 0 and cur > p.min_replicas:
                target = cur - 1
            if target != cur:
                log.info("planner: %s %d -> %d (kv=%.2f queue=%.1f)",
                         comp, cur, target, kv, qd)
                await self.connector.scale(comp, target)
                self.last_action[comp] = now


@dataclass
class SLATargets:
    ttft_s: float = 2.0
    itl_ms: float = 25.0
    isl: int = 8192
    osl: int = 1024


@dataclass
class PerfModel:
    """Measured single-replica capability (profiler output — the analog of
    the reference's pre-deployment SLA profiling, profile_sla.py:343)."""
    prefill_tokens_per_s: float = 100_000.0   # per prefill replica
    decode_tokens_per_s_at_itl: float = 600.0  # per decode replica at SLA ITL
    max_conc_at_itl: int = 16


class ThroughputPlanner:
    """SLA-driven replica computation from predicted request rate
    (core/throughput_scaling.py parity: compute prefill/decode replicas)."""

    def __init__(self, sla: SLATargets, perf: PerfModel, connector: Connector,
                 predictor: str = "moving_average",
                 prefill_component: str = "prefill",
                 decode_component: str = "backend",
                 max_replicas: int = 64):
        self.sla = sla
        self.perf = perf
        self.connector = connector
        self.rate_pred = PREDICTORS[predictor]()
        self.prefill_component = prefill_component
        self.decode_component = decode_component
        self.max_replicas = max_replicas

    def required_replicas(self, req_per_s: float) -> Dict[str, int]:
        prefill_load = req_per_s * self.sla.isl
        n_prefill = max(1, math.ceil(
            prefill_load / self.perf.prefill_tokens_per_s))
        # each in-flight request occupies a decode slot for osl * itl
        inflight = req_per_s * self.sla.osl * (self.sla.itl_ms / 1000.0)
        n_decode = max(1, math.ceil(inflight / self.perf.max_conc_at_itl))
        return {self.prefill_component: min(self.max_replicas, n_prefill),
                self.decode_component: min(self.max_replicas, n_decode)}

    async def observe_and_plan(self, req_per_s: float):
        self.rate_pred.observe(req_per_s)
        targets = self.required_replicas(self.rate_pred.predict())
        for comp, n in targets.items():
            if self.connector.current(comp) != n:
                await self.connector.scale(comp, n)
        return targets


class PlannerService:
    """Polls worker metrics via the request plane and drives a LoadPlanner
    (the deployment loop of `python -m dynamo.planner`)."""

    def __init__(self, runtime, namespace: str, planner: LoadPlanner,
                 interval: float = 2.0):
        self.runtime = runtime
        self.namespace = namespace
        self.planner = planner
        self.interval = interval
        self._task: Optional[asyncio.Task] = None

    async def start(self):
        self._task = asyncio.create_task(self._loop())
        return self

    async def stop(self):
        if self._task:
            self._task.cancel()

    async def _loop(self):
        while True:
            try:
                obs: Dict[str, PoolObservation] = {}
                for comp in self.planner.policies:
                    insts = self.runtime.discovery.list(self.namespace, comp)
                    if not insts:
                        continue
                    o = PoolObservation(replicas=len(insts))
                    for inst in insts:
                        try:
                            m = await self.runtime.client.call(
                                inst.address, f"{comp}.get_perf_metrics", {})
                            o.kv_usage = max(o.kv_usage, m.get("kv_usage", 0))
                            o.num_waiting += m.get("num_waiting", 0)
                            o.num_running += m.get("num_running", 0)
                        except Exception:
                            pass
                    obs[comp] = o
                await self.planner.observe_and_plan(obs)
            except Exception:
                log.exception("planner loop failed")
            await asyncio.sleep(self.interval)
