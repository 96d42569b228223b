"""Discovery plane: lease-based instance registry with watchers.

Backends (mirroring the reference's pluggable discovery,
ai-dynamo/dynamo lib/runtime/src/discovery/{kv_store.rs,mock.rs} and the
etcd path scheme /services/{ns}/{comp}/{ep}-{lease_id}):
  - MemoryDiscovery: in-process (tests, single-process deployments)
  - FileDiscovery:   shared-directory JSON files with mtime leases (no etcd
    in this environment; a node-local control plane needs no quorum store)

An Instance record carries the worker's address, endpoint names, and the
model card (the worker -> frontend contract, model_card.rs:834 parity).
"""
from __future__ import annotations

import json
import os
import time
import uuid
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional


@dataclass
class Instance:
    namespace: str
    component: str
    instance_id: str
    address: str                       # "host:port" on the request plane
    endpoints: List[str] = field(default_factory=list)
    model_card: Optional[dict] = None
    metadata: dict = field(default_factory=dict)
    registered_at: float = field(default_factory=time.time)

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.component}/{self.instance_id}"

    def to_dict(self) -> dict:
        return {
            "namespace": self.namespace, "component": self.component,
            "instance_id": self.instance_id, "address": self.address,
            "endpoints": self.endpoints, "model_card": self.model_card,
            "metadata": self.metadata, "registered_at": self.registered_at,
        }

    @staticmethod
    def from_dict(d: dict) -> "Instance":
        return Instance(**d)


class DiscoveryBackend:
    """register/refresh/deregister + list/watch."""

    def register(self, inst: Instance) -> None:
        raise NotImplementedError

    def refresh(self, inst: Instance) -> None:
        raise NotImplementedError

    def deregister(self, inst: Instance) -> None:
        raise NotImplementedError

    def list(self, namespace: str, component: Optional[str] = None) -> List[Instance]:
        raise NotImplementedError


class MemoryDiscovery(DiscoveryBackend):
    def __init__(self):
        self._instances: Dict[str, Instance] = {}

    def register(self, inst: Instance):
        self._instances[inst.key] = inst

    def refresh(self, inst: Instance):
        self._instances[inst.key] = inst

    def deregister(self, inst: Instance):
        self._instances.pop(inst.key, None)

    def list(self, namespace, component=None):
        return [i for i in self._instances.values()
                if i.namespace == namespace
                and (component is None or i.component == component)]


class FileDiscovery(DiscoveryBackend):
    """Lease = file mtime; instances older than ttl are dead (the etcd
    lease keep-alive analog, transports/etcd/lease.rs)."""

    def __init__(self, root: str, ttl: float = 10.0):
        self.root = root
        self.ttl = ttl
        os.makedirs(root, exist_ok=True)

    def _path(self, inst: Instance) -> str:
        d = os.path.join(self.root, inst.namespace, inst.component)
        os.makedirs(d, exist_ok=True)
        return os.path.join(d, inst.instance_id + ".json")

    def register(self, inst: Instance):
        p = self._path(inst)
        tmp = p + ".tmp"
        with open(tmp, "w") as f:
            json.dump(inst.to_dict(), f)
        os.replace(tmp, p)

    def refresh(self, inst: Instance):
        p = self._path(inst)
        if os.path.exists(p):
            os.utime(p)
        else:
            self.register(inst)

    def deregister(self, inst: Instance):
        try:
            os.remove(self._path(inst))
        except FileNotFoundError:
            pass

    def list(self, namespace, component=None):
        out = []
        nsdir = os.path.join(self.root, namespace)
        if not os.path.isdir(nsdir):
            return out
        comps = [component] if component else os.listdir(nsdir)
        now = time.time()
        for c in comps:
            cdir = os.path.join(nsdir, c)
            if not os.path.isdir(cdir):
                continue
            for fn in os.listdir(cdir):
                if not fn.endswith(".json"):
                    continue
                p = os.path.join(cdir, fn)
                try:
                    if now - os.path.getmtime(p) > self.ttl:
                        continue  # lease expired
                    with open(p) as f:
                        out.append(Instance.from_dict(json.load(f)))
                except (OSError, json.JSONDecodeError):
                    continue
        return out


def make_discovery(spec: str) -> DiscoveryBackend:
    """spec: "memory" or "file:/path" (DYN_DISCOVERY_BACKEND analog)."""
    if spec == "memory":
        return MemoryDiscovery()
    if spec.startswith("file:"):
        return FileDiscovery(spec[5:])
    raise ValueError(f"unknown discovery backend {spec!r}")


def new_instance_id() -> str:
    return uuid.uuid4().hex[:12]
