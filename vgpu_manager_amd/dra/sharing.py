"""Sharing-strategy configs for DRA claims (reference
pkg/kubeletplugin/sharing.go, which maps MPS / TimeSlicing configs to
per-claim CUDA envs).

MI355X has no MPS daemon; the native equivalents are:

  * ``time-slicing``  — the CU token-bucket throttle: consumers get a
    core_limit share (equal split or explicit per-consumer percents)
    enforced by the in-container shim — the temporal analog of
    NVIDIA's time-sliced scheduling.
  * ``cu-partition``  — spatial sharing on CDNA4: consumers are pinned
    to disjoint CPX partitions (one or more XCDs each, 32 CUs per
    XCD), the structural analog of MPS active-thread-percentage.
    Emitted as partition indices the CDI edits translate to
    render-node visibility.

A sharing config arrives as the claim's opaque config parameters:
``{"strategy": "time-slicing"|"cu-partition"|"none", ...}``.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .state import CPX_PARTITIONS_PER_GPU, VgpuClaimParams

VALID_STRATEGIES = ("none", "time-slicing", "cu-partition")


class SharingError(Exception):
    pass


@dataclass
class SharingDecision:
    """What prepare() applies on top of the raw claim params.
    Keyed by consumer index (several consumers can share one GPU UUID
    — that is the whole point of sharing)."""
    strategy: str = "none"
    # consumer index -> core_limit override (time-slicing)
    core_limits: Dict[int, int] = field(default_factory=dict)
    # consumer index -> list of CPX partition indices (cu-partition)
    partitions: Dict[int, List[int]] = field(default_factory=dict)


def apply_sharing_config(params: List[VgpuClaimParams],
                         config: Optional[dict]) -> SharingDecision:
    """Validate + resolve a sharing config against the claim devices.

    time-slicing: `slices` (default len(params)) equal shares, or
    `percents` (list, must sum <=100, one per consumer in order).
    cu-partition: `partitionsPerConsumer` (default 8/len(params),
    i.e. the GPU's XCDs split evenly) disjoint CPX partitions.
    """
    dec = SharingDecision()
    if not config:
        return dec
    strategy = config.get("strategy", "none")
    if strategy not in VALID_STRATEGIES:
        raise SharingError(f"unknown sharing strategy {strategy!r}")
    dec.strategy = strategy
    if strategy == "none" or not params:
        return dec

    if strategy == "time-slicing":
        percents = config.get("percents")
        if percents is not None:
            if len(percents) != len(params):
                raise SharingError(
                    f"percents has {len(percents)} entries for "
                    f"{len(params)} consumers")
            if any(not 0 < int(p) <= 100 for p in percents):
                raise SharingError("percents entries must be in 1..100")
            if sum(int(p) for p in percents) > 100:
                raise SharingError("percents sum exceeds 100")
            for i, pc in enumerate(percents):
                dec.core_limits[i] = int(pc)
        else:
            slices = int(config.get("slices", len(params)))
            if slices < 1:
                raise SharingError("slices must be >= 1")
            share = max(1, 100 // slices)
            for i in range(len(params)):
                dec.core_limits[i] = share
        return dec

    # cu-partition: hand out disjoint CPX partitions round-robin
    per = config.get("partitionsPerConsumer")
    n = len(params)
    if per is None:
        per = max(1, CPX_PARTITIONS_PER_GPU // n)
    per = int(per)
    if per < 1 or per * n > CPX_PARTITIONS_PER_GPU:
        raise SharingError(
            f"{n} consumers x {per} partitions exceeds "
            f"{CPX_PARTITIONS_PER_GPU} XCDs")
    nxt = 0
    for i in range(n):
        dec.partitions[i] = list(range(nxt, nxt + per))
        nxt += per
    return dec
