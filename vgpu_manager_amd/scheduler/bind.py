"""Bind verb: verify the pre-allocation targets this node, create the
Binding (reference pkg/scheduler/bind/bind_predicate.go:106)."""
from __future__ import annotations

import logging
import time

from ..client.kube import KubeClient, KubeError
from ..util import consts
from . import metrics
from .serial import KeyedLocker

log = logging.getLogger("vgpu.scheduler.bind")


class NodeBinder:
    def __init__(self, client: KubeClient, serialize: bool = True):
        self.client = client
        self.locker = KeyedLocker() if serialize else None

    def bind(self, args: dict) -> dict:
        t0 = time.monotonic()
        name = args.get("PodName") or args.get("podName")
        namespace = args.get("PodNamespace") or args.get("podNamespace") \
            or "default"
        node = args.get("Node") or args.get("node")
        key = f"{namespace}/{name}"
        if self.locker:
            self.locker.acquire(key)
        try:
            try:
                pod = self.client.get_pod(namespace, name)
            except KubeError as e:
                return {"Error": str(e)}
            ann = pod.get("metadata", {}).get("annotations", {}) or {}
            predicate = ann.get(consts.predicate_node_ann())
            if not ann.get(consts.pre_alloc_ann()):
                return {"Error":
                        f"pod {key} has no pre-allocated devices"}
            if predicate and predicate != node:
                return {"Error":
                        f"pre-allocated node {predicate} != bind target "
                        f"{node}"}
            try:
                self.client.create_binding(namespace, name, node)
            except KubeError as e:
                metrics.observe("bind", time.monotonic() - t0, False)
                return {"Error": str(e)}
            metrics.observe("bind", time.monotonic() - t0, True)
            return {"Error": ""}
        finally:
            if self.locker:
                self.locker.release(key)
