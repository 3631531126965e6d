"""Scheduler stuck detection (reference stuck-grace-period +
MustInitGlobalStuckGracePeriod, cmd/device-monitor/main.go:67):
pods pre-allocated but never bound past the grace period are surfaced
as events/metrics and optionally have their pre-allocation cleared so
the scheduler retries cleanly."""
from __future__ import annotations

import logging
import time
from typing import List

from ..client.kube import KubeClient, KubeError
from ..util import consts

log = logging.getLogger("vgpu.monitor.stuck")

DEFAULT_GRACE_S = 120


def find_stuck_pods(client: KubeClient,
                    default_grace_s: int = DEFAULT_GRACE_S) -> List[dict]:
    try:
        pods = client.list_pods(
            label_selector={consts.assigned_phase_label():
                            consts.PHASE_ALLOCATING})
    except KubeError:
        return []
    now = time.time()
    stuck = []
    for p in pods:
        if p.get("spec", {}).get("nodeName"):
            continue  # bound; kubelet will progress it
        ann = p.get("metadata", {}).get("annotations", {}) or {}
        t = ann.get(consts.predicate_time_ann())
        if not t:
            continue
        grace = default_grace_s
        g = ann.get(consts.stuck_grace_period_ann())
        if g:
            try:
                grace = int(g)
            except ValueError:
                pass
        try:
            if now - int(t) > grace:
                stuck.append(p)
        except ValueError:
            continue
    return stuck


def recover_stuck_pods(client: KubeClient, *, clear: bool = True,
                       default_grace_s: int = DEFAULT_GRACE_S) -> int:
    n = 0
    for p in find_stuck_pods(client, default_grace_s):
        meta = p.get("metadata", {})
        ns, name = meta.get("namespace", "default"), meta.get("name", "")
        log.warning("stuck pod %s/%s (pre-allocated, never bound)", ns,
                    name)
        client.create_event(
            ns, {"kind": "Pod", "name": name, "namespace": ns,
                 "uid": meta.get("uid", "")},
            "VGPUSchedulingStuck",
            "pre-allocated but not bound within grace period")
        if clear:
            try:
                client.patch_pod_metadata(
                    ns, name,
                    annotations={consts.pre_alloc_ann(): "",
                                 consts.predicate_node_ann(): "",
                                 consts.predicate_time_ann(): ""},
                    labels={consts.assigned_phase_label(): ""})
            except KubeError as e:
                log.error("clearing stuck pod failed: %s", e)
                continue
        n += 1
    return n
