"""Global planner: centralized scale execution for multiple pools.

Reference parity: components/src/dynamo/global_planner — local planners
delegate replica-scale execution to one central service which enforces
shared constraints (total GPU budget, per-pool min/max) across pools.
Here the execution side applies decisions through per-pool Connectors
(VirtualConnector in tests, SubprocessConnector in deployments), and the
service side exposes a `scale` endpoint on the request plane that a
GlobalPlannerConnector (used as a local planner's connector) delegates to.
"""
from __future__ import annotations

import asyncio
import logging
import math
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from .planner import Connector

log = logging.getLogger("dynamo_amd.global_planner")


@dataclass
class PoolBudgetPolicy:
    pool: str
    weight: float = 1.0          # fair-share weight under contention
    min_replicas: int = 1        # guaranteed floor (per component total)
    max_replicas: int = 64


class GlobalPlanner:
    """Arbitrates replica requests from local planners under a shared
    total budget. Grants are per (pool, component); under contention the
    surplus above each pool's floor is distributed by weighted fair share."""

    def __init__(self, total_budget: int,
                 policies: List[PoolBudgetPolicy],
                 executors: Optional[Dict[str, Connector]] = None):
        self.total_budget = total_budget
        self.policies = {p.pool: p for p in policies}
        self.executors = executors or {}
        # desired replicas per (pool, component) as requested by local
        # planners; grants computed from these
        self.requested: Dict[Tuple[str, str], int] = {}
        self.granted: Dict[Tuple[str, str], int] = {}
        self.history: List[tuple] = []

    def _pool_requested(self, pool: str) -> int:
        return sum(n for (p, _c), n in self.requested.items() if p == pool)

    def _arbitrate(self) -> Dict[Tuple[str, str], int]:
        """Compute grants. Floors first; remaining budget split by weight;
        a pool never gets more than it asked for (surplus redistributes)."""
        pools = {p for (p, _c) in self.requested}
        want = {p: max(self._pool_requested(p),
                       self.policies.get(p, PoolBudgetPolicy(p)).min_replicas)
                for p in pools}
        want = {p: min(v, self.policies.get(p, PoolBudgetPolicy(p)).max_replicas)
                for p, v in want.items()}
        if sum(want.values()) <= self.total_budget:
            alloc = dict(want)
        else:
            alloc = {p: min(want[p],
                            self.policies.get(p, PoolBudgetPolicy(p)).min_replicas)
                     for p in pools}
            remaining = self.total_budget - sum(alloc.values())
            # weighted fair share of the remainder, capped at demand
            live = {p for p in pools if alloc[p] < want[p]}
            while remaining > 0 and live:
                wsum = sum(self.policies.get(p, PoolBudgetPolicy(p)).weight
                           for p in live)
                progress = False
                for p in sorted(live,
                                key=lambda q: -self.policies.get(
                                    q, PoolBudgetPolicy(q)).weight):
                    share = max(1, math.floor(
                        remaining * self.policies.get(
                            p, PoolBudgetPolicy(p)).weight / max(wsum, 1e-9)))
                    take = min(share, want[p] - alloc[p], remaining)
                    if take > 0:
                        alloc[p] += take
                        remaining -= take
                        progress = True
                    if alloc[p] >= want[p]:
                        live.discard(p)
                if not progress:
                    break
        # distribute each pool's allocation over its components
        # proportionally to the requested split
        grants: Dict[Tuple[str, str], int] = {}
        for p in pools:
            comps = [(c, n) for (pp, c), n in self.requested.items()
                     if pp == p]
            total_req = sum(n for _c, n in comps) or 1
            budget = alloc.get(p, 0)
            left = budget
            for i, (c, n) in enumerate(sorted(comps)):
                if i == len(comps) - 1:
                    g = left
                else:
                    g = min(left, max(1 if n > 0 else 0,
                                      math.floor(budget * n / total_req)))
                grants[(p, c)] = min(g, n)
                left -= grants[(p, c)]
        return grants

    async def request_scale(self, pool: str, component: str,
                            target: int) -> int:
        """A local planner asks for `target` replicas of (pool, component);
        returns the granted count after budget arbitration and applies it
        through the pool's executor."""
        self.requested[(pool, component)] = max(0, int(target))
        grants = self._arbitrate()
        changed = []
        for key, g in grants.items():
            if self.granted.get(key) != g:
                self.granted[key] = g
                changed.append((key, g))
        self.history.append((time.time(), pool, component, target,
                             grants.get((pool, component), 0)))
        for (p, c), g in changed:
            ex = self.executors.get(p)
            if ex is not None:
                await ex.scale(c, g)
        return grants.get((pool, component), 0)

    @property
    def used(self) -> int:
        return sum(self.granted.values())


class GlobalPlannerService:
    """Serves `scale` + `status` on the request plane (component
    `global_planner`) so local planners in other processes can delegate."""

    def __init__(self, runtime, planner: GlobalPlanner,
                 namespace: str = "dynamo"):
        self.runtime = runtime
        self.planner = planner
        self.comp = runtime.namespace(namespace).component("global_planner")

    async def start(self):
        self.comp.serve_endpoint("scale", self._scale)
        self.comp.serve_endpoint("status", self._status)
        await self.comp.register(
            model_card=None, metadata={"role": "global_planner",
                                       "budget": self.planner.total_budget})
        return self

    async def stop(self):
        self.comp.deregister()

    async def _scale(self, payload, ctx):
        granted = await self.planner.request_scale(
            payload["pool"], payload["component"], int(payload["target"]))
        yield {"granted": granted, "used": self.planner.used,
               "budget": self.planner.total_budget}

    async def _status(self, payload, ctx):
        yield {"budget": self.planner.total_budget,
               "used": self.planner.used,
               "granted": {f"{p}/{c}": n
                           for (p, c), n in self.planner.granted.items()}}


class GlobalPlannerConnector(Connector):
    """A local planner's Connector that delegates to a GlobalPlannerService
    (reference: planner connectors/global_planner.py). The local planner
    keeps its own policy; the grant (possibly below target) is what takes
    effect locally."""

    def __init__(self, runtime, pool: str, namespace: str = "dynamo",
                 local: Optional[Connector] = None):
        self.runtime = runtime
        self.pool = pool
        self.namespace = namespace
        self.local = local            # applies the granted count locally
        self._granted: Dict[str, int] = {}

    async def scale(self, component: str, target: int):
        insts = self.runtime.discovery.list(self.namespace, "global_planner")
        if not insts:
            raise RuntimeError("no global planner registered")
        r = await self.runtime.client.call(
            insts[0].address, "global_planner.scale",
            {"pool": self.pool, "component": component, "target": target})
        granted = int(r["granted"])
        self._granted[component] = granted
        if self.local is not None:
            await self.local.scale(component, granted)
        return granted

    def current(self, component: str) -> int:
        if self.local is not None:
            return self.local.current(component)
        return self._granted.get(component, 0)
