"""DistributedRuntime -> Namespace -> Component -> Endpoint hierarchy.

The orchestration object model of the reference's runtime
(ai-dynamo/dynamo lib/runtime/src/distributed.rs:53 DistributedRuntime,
component.rs:178/361/470 Component/Endpoint/Namespace): a worker process
creates a runtime, registers its component instance (with model card) into
discovery under a lease, and serves endpoints on the request plane; clients
watch discovery and push requests with a selectable router mode
(push_router.rs:209 RouterMode).
"""
from __future__ import annotations

import asyncio
import random
import time
from typing import Any, AsyncIterator, Dict, List, Optional

from .discovery import (DiscoveryBackend, Instance, make_discovery,
                        new_instance_id)
from .request_plane import (EndpointError, Handler, RequestPlaneClient,
                            RequestPlaneServer)


class DistributedRuntime:
    def __init__(self, discovery: "DiscoveryBackend | str" = "memory",
                 host: str = "127.0.0.1", lease_ttl: float = 10.0):
        self.discovery = (discovery if isinstance(discovery, DiscoveryBackend)
                          else make_discovery(discovery))
        self.host = host
        self.lease_ttl = lease_ttl
        self.server = RequestPlaneServer(host)
        self.client = RequestPlaneClient()
        self._instances: List[Instance] = []
        self._lease_task: Optional[asyncio.Task] = None
        self._started = False

    def namespace(self, name: str) -> "Namespace":
        return Namespace(self, name)

    async def start(self):
        if not self._started:
            await self.server.start()
            self._started = True
            self._lease_task = asyncio.create_task(self._lease_loop())
        return self

    async def _lease_loop(self):
        period = max(0.5, self.lease_ttl / 3)
        while True:
            await asyncio.sleep(period)
            for inst in self._instances:
                try:
                    self.discovery.refresh(inst)
                except Exception:
                    pass

    async def shutdown(self, drain: bool = True):
        if self._lease_task:
            self._lease_task.cancel()
        for inst in self._instances:
            try:
                self.discovery.deregister(inst)
            except Exception:
                pass
        await self.server.stop(drain=drain)
        await self.client.close()

    def register_instance(self, inst: Instance):
        self.discovery.register(inst)
        self._instances.append(inst)


class Namespace:
    def __init__(self, runtime: DistributedRuntime, name: str):
        self.runtime = runtime
        self.name = name

    def component(self, name: str) -> "Component":
        return Component(self, name)


class Component:
    def __init__(self, ns: Namespace, name: str):
        self.ns = ns
        self.name = name
        self.runtime = ns.runtime
        self.instance_id = new_instance_id()
        self._endpoints: List[str] = []
        self._instance: Optional[Instance] = None

    def endpoint(self, name: str) -> "Endpoint":
        return Endpoint(self, name)

    async def register(self, model_card: Optional[dict] = None,
                       metadata: Optional[dict] = None):
        """Publish this component instance (and its model card) to
        discovery — register_model parity (bindings rust/lib.rs:186)."""
        await self.runtime.start()
        self._instance = Instance(
            namespace=self.ns.name, component=self.name,
            instance_id=self.instance_id, address=self.runtime.server.address,
            endpoints=list(self._endpoints), model_card=model_card,
            metadata=metadata or {})
        self.runtime.register_instance(self._instance)
        return self._instance

    def update_metadata(self, **kw):
        if self._instance:
            self._instance.metadata.update(kw)
            self.runtime.discovery.refresh(self._instance)

    def serve_endpoint(self, name: str, handler: Handler):
        """Register handler under '{component}.{name}'."""
        full = f"{self.name}.{name}"
        self.runtime.server.add_endpoint(full, handler)
        self._endpoints.append(name)

    def deregister(self):
        if self._instance:
            self.runtime.discovery.deregister(self._instance)


class Endpoint:
    def __init__(self, component: Component, name: str):
        self.component = component
        self.name = name

    def client(self, **kw) -> "PushClient":
        return PushClient(self.component.runtime,
                          self.component.ns.name, self.component.name,
                          self.name, **kw)


class NoInstancesError(RuntimeError):
    pass


class PushClient:
    """Client-side router over the live instances of a component endpoint.

    Modes mirror push_router.rs:209: round_robin, random, direct
    (instance_id); kv-aware and load-based selection live in
    dynamo_amd.router on top of this.
    """

    def __init__(self, runtime: DistributedRuntime, namespace: str,
                 component: str, endpoint: str, mode: str = "round_robin"):
        self.runtime = runtime
        self.namespace = namespace
        self.component = component
        self.endpoint = endpoint
        self.mode = mode
        self._rr = 0
        # local inhibition of recently-failed instances (distributed-runtime
        # "Local Worker Inhibition")
        self._inhibited: Dict[str, float] = {}
        self.inhibit_secs = 5.0

    def instances(self) -> List[Instance]:
        now = time.time()
        out = []
        for inst in self.runtime.discovery.list(self.namespace, self.component):
            if self.endpoint not in inst.endpoints:
                continue
            if self._inhibited.get(inst.instance_id, 0) > now:
                continue
            out.append(inst)
        return sorted(out, key=lambda i: i.instance_id)

    def inhibit(self, instance_id: str):
        self._inhibited[instance_id] = time.time() + self.inhibit_secs

    def pick(self, instance_id: Optional[str] = None) -> Instance:
        insts = self.instances()
        if not insts:
            raise NoInstancesError(
                f"no live instances for {self.namespace}/{self.component}"
                f".{self.endpoint}")
        if instance_id is not None:  # direct mode
            for i in insts:
                if i.instance_id == instance_id:
                    return i
            raise NoInstancesError(f"instance {instance_id} not found")
        if self.mode == "random":
            return random.choice(insts)
        self._rr += 1
        return insts[self._rr % len(insts)]

    async def generate(self, payload: Any,
                       instance_id: Optional[str] = None) -> AsyncIterator[Any]:
        inst = self.pick(instance_id)
        full = f"{self.component}.{self.endpoint}"
        try:
            async for chunk in self.runtime.client.call_stream(
                    inst.address, full, payload):
                yield chunk
        except (ConnectionRefusedError, ConnectionResetError, OSError) as e:
            self.inhibit(inst.instance_id)
            raise EndpointError(f"worker {inst.instance_id} unreachable: {e}")

    async def call(self, payload: Any, instance_id: Optional[str] = None) -> Any:
        last = None
        async for chunk in self.generate(payload, instance_id):
            last = chunk
        return last
