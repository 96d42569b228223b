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


class ARIMAPredictor(Predictor):
    """AR(p) on first differences (ARIMA(p,1,0)) fitted by least squares —
    the ARIMA-class predictor of the reference's planner family
    (components/src/dynamo/planner core/load/predictors.py,
    simulation/load_predictor.py) without a stats dependency. Falls back
    to a linear trend until enough history accumulates."""

    def __init__(self, window: int = 48, p: int = 4):
        self.buf: Deque[float] = deque(maxlen=window)
        self.p = p
        self._trend = TrendPredictor(window=min(window, 12))

    def observe(self, v):
        self.buf.append(v)
        self._trend.observe(v)

    def predict(self):
        n = len(self.buf)
        if n < self.p + 4:
            return self._trend.predict()
        import numpy as np
        y = np.asarray(self.buf, dtype=np.float64)
        d = np.diff(y)                       # first differences
        p = self.p
        X = np.column_stack([d[i:len(d) - p + i] for i in range(p)])
        t = d[p:]
        try:
            coef, *_ = np.linalg.lstsq(X, t, rcond=None)
        except Exception:
            return self._trend.predict()
        nxt = float(np.dot(coef, d[-p:]))    # next difference
        return max(0.0, y[-1] + nxt)


PREDICTORS = {"constant": ConstantPredictor, "moving_average":
              MovingAveragePredictor, "trend": TrendPredictor,
              "arima": ARIMAPredictor}


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
class ParallelizationConfig:
    """Per-pool parallelism mapping (reference parity:
    components/src/dynamo/planner/config/parallelization.py:99-126):
    how many GPUs one replica of each worker type occupies. The planner
    scales REPLICAS; deployments budget GPUS - this is the conversion,
    plus the cluster-cap arbitration applied to scaling decisions."""
    tp_size: int = 1
    pp_size: int = 1
    moe_tp_size: int = 0          # 0 = dense model (use tp_size)
    moe_ep_size: int = 0
    dp_size: int = 1

    def __post_init__(self):
        for f in ("tp_size", "pp_size", "dp_size"):
            if getattr(self, f) < 1:
                raise ValueError(f"{f} must be >= 1")
        if (self.moe_tp_size > 0) != (self.moe_ep_size > 0):
            raise ValueError("moe_tp_size and moe_ep_size go together")

    @property
    def gpus_per_replica(self) -> int:
        attn = self.tp_size
        moe = (self.moe_tp_size * self.moe_ep_size
               if self.moe_tp_size else attn)
        return max(attn, moe) * self.pp_size * self.dp_size


def cap_to_gpu_budget(targets: Dict[str, int],
                      parallel: "Dict[str, ParallelizationConfig]",
                      total_gpus: int) -> Dict[str, int]:
    """Scale replica targets down proportionally when the plan exceeds the
    cluster GPU budget, never below 1 replica per pool (the reference
    planner's budget arbitration step)."""
    if total_gpus <= 0:
        return dict(targets)
    cost = {c: parallel.get(c, ParallelizationConfig()).gpus_per_replica
            for c in targets}
    need = sum(targets[c] * cost[c] for c in targets)
    if need <= total_gpus:
        return dict(targets)
    scale = total_gpus / need
    capped = {c: max(1, int(targets[c] * scale)) for c in targets}
    while sum(capped[c] * cost[c] for c in capped) > total_gpus:
        big = max((c for c in capped if capped[c] > 1),
                  key=lambda c: capped[c] * cost[c], default=None)
        if big is None:
            break
        capped[big] -= 1
    return capped


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


def _interp(x: float, xs: List[float], ys: List[float]) -> float:
    """Piecewise-linear interpolation with flat extrapolation."""
    if not xs:
        return 0.0
    if x <= xs[0]:
        return ys[0]
    if x >= xs[-1]:
        return ys[-1]
    for i in range(1, len(xs)):
        if x <= xs[i]:
            f = (x - xs[i - 1]) / (xs[i] - xs[i - 1])
            return ys[i - 1] + f * (ys[i] - ys[i - 1])
    return ys[-1]


class InterpolatedPerfModel:
    """Perf model interpolated from a profiler concurrency sweep
    (reference parity: planner core/perf_model/{prefill,decode}.py + AIC
    interpolation monitoring/aic_*.py — measured (concurrency, ITL,
    throughput) points instead of a single operating point)."""

    def __init__(self, sweep: List[dict], isl: int):
        pts = sorted((r for r in sweep if r.get("itl_p50_ms")),
                     key=lambda r: r["concurrency"])
        self.conc = [float(r["concurrency"]) for r in pts]
        self.itl = [float(r["itl_p50_ms"]) for r in pts]
        self.tps = [float(r.get("output_tok_s", 0.0)) for r in pts]
        ttfts = [r.get("ttft_p50_s") for r in pts if r.get("ttft_p50_s")]
        # prefill rate from the least-loaded level's TTFT
        self.prefill_tokens_per_s = (isl / ttfts[0]) if ttfts else 1e9
        self.isl = isl

    @classmethod
    def from_profile(cls, profile: dict) -> "InterpolatedPerfModel":
        """Build from `python -m dynamo_amd.profiler` output JSON."""
        return cls(profile["sweep"], int(profile.get("isl", 8192)))

    def max_conc_at_itl(self, itl_slo_ms: float) -> float:
        """Largest per-replica concurrency whose interpolated ITL meets
        the SLO (ITL grows monotonically with concurrency)."""
        if not self.conc:
            return 1.0
        if self.itl[-1] <= itl_slo_ms:
            return self.conc[-1]
        if self.itl[0] > itl_slo_ms:
            return max(1.0, self.conc[0] * itl_slo_ms / self.itl[0])
        # invert the (conc -> itl) curve at the SLO
        return _interp(itl_slo_ms, self.itl, self.conc)

    def decode_tps_at(self, conc: float) -> float:
        return _interp(conc, self.conc, self.tps)


class CorrectionFactors:
    """EWMA observed/expected ratios applied to the replica computation
    (reference: throughput_scaling.py correction factors): if measured
    ITL or TTFT runs hotter than the perf model predicts at the current
    load, requirements inflate proportionally."""

    def __init__(self, alpha: float = 0.3, lo: float = 0.25, hi: float = 4.0):
        self.alpha = alpha
        self.lo, self.hi = lo, hi
        self.prefill = 1.0
        self.decode = 1.0

    def _upd(self, cur: float, ratio: float) -> float:
        ratio = min(self.hi, max(self.lo, ratio))
        return (1 - self.alpha) * cur + self.alpha * ratio

    def observe(self, expected_itl_ms: float, actual_itl_ms: Optional[float],
                expected_ttft_s: float, actual_ttft_s: Optional[float]):
        if actual_itl_ms and expected_itl_ms > 0:
            self.decode = self._upd(self.decode,
                                    actual_itl_ms / expected_itl_ms)
        if actual_ttft_s and expected_ttft_s > 0:
            self.prefill = self._upd(self.prefill,
                                     actual_ttft_s / expected_ttft_s)


class ScalingState:
    STEADY = "steady"
    SCALE_UP = "scale_up"
    SCALE_DOWN = "scale_down"
    COOLDOWN = "cooldown"


class SLAPlanner:
    """Closed-loop SLA planner: predicted request rate (ARIMA-class) ->
    interpolated perf model + correction factors -> replica targets, with
    an explicit scaling state machine (reference core/state_machine.py):
    scale-up is immediate, scale-down requires `down_stable` consecutive
    under-loaded intervals, and every action enters COOLDOWN."""

    def __init__(self, sla: SLATargets, perf: InterpolatedPerfModel,
                 connector: Connector, predictor: str = "arima",
                 prefill_component: str = "prefill",
                 decode_component: str = "backend",
                 max_replicas: int = 64, cooldown_s: float = 10.0,
                 down_stable: int = 3,
                 parallel: "Optional[Dict[str, ParallelizationConfig]]" = None,
                 total_gpus: int = 0):
        self.sla = sla
        self.perf = perf
        self.connector = connector
        self.rate_pred = PREDICTORS[predictor]()
        self.corrections = CorrectionFactors()
        self.prefill_component = prefill_component
        self.decode_component = decode_component
        self.max_replicas = max_replicas
        self.cooldown_s = cooldown_s
        self.down_stable = down_stable
        self.parallel = parallel
        self.total_gpus = total_gpus
        self.state = ScalingState.STEADY
        self._last_action = 0.0
        self._down_count: Dict[str, int] = {}

    def required_replicas(self, req_per_s: float) -> Dict[str, int]:
        c = self.corrections
        prefill_load = req_per_s * self.sla.isl * c.prefill
        n_prefill = max(1, math.ceil(
            prefill_load / max(1.0, self.perf.prefill_tokens_per_s)))
        # per-replica concurrency the ITL SLO allows, deflated by the
        # decode correction (hotter-than-modeled => fewer slots/replica)
        conc = max(1.0, self.perf.max_conc_at_itl(self.sla.itl_ms) / c.decode)
        tps = max(1.0, self.perf.decode_tps_at(conc))
        # steady-state decode demand: each request needs osl tokens over
        # its lifetime; a replica sustains `tps` output tokens/s at SLA
        n_decode = max(1, math.ceil(req_per_s * self.sla.osl / tps))
        targets = {self.prefill_component: min(self.max_replicas, n_prefill),
                   self.decode_component: min(self.max_replicas, n_decode)}
        if self.parallel is not None:
            targets = cap_to_gpu_budget(targets, self.parallel,
                                        self.total_gpus)
        return targets

    async def observe_and_plan(self, req_per_s: float,
                               actual_itl_ms: Optional[float] = None,
                               actual_ttft_s: Optional[float] = None
                               ) -> Dict[str, int]:
        self.rate_pred.observe(req_per_s)
        self.corrections.observe(self.sla.itl_ms, actual_itl_ms,
                                 self.sla.ttft_s, actual_ttft_s)
        targets = self.required_replicas(self.rate_pred.predict())
        now = time.time()
        if now - self._last_action < self.cooldown_s:
            self.state = ScalingState.COOLDOWN
            return targets
        acted = False
        for comp, n in targets.items():
            cur = self.connector.current(comp)
            if n > cur:
                self._down_count[comp] = 0
                await self.connector.scale(comp, n)
                self.state = ScalingState.SCALE_UP
                acted = True
            elif n < cur:
                # scale-down needs sustained under-load (state machine:
                # avoid flapping on a transient dip)
                self._down_count[comp] = self._down_count.get(comp, 0) + 1
                if self._down_count[comp] >= self.down_stable:
                    await self.connector.scale(comp, n)
                    self.state = ScalingState.SCALE_DOWN
                    self._down_count[comp] = 0
                    acted = True
            else:
                self._down_count[comp] = 0
        if acted:
            self._last_action = now
        elif self.state != ScalingState.COOLDOWN:
            self.state = ScalingState.STEADY
        return targets


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
