"""Region supervisor: heartbeats → φ-accrual → failover migration.

Reference parity: src/meta-srv/src/region/supervisor.rs (RegionSupervisor
runs the φ detector per datanode and generates region-migration failover
tasks) + datanode alive_keeper leases. MI355X mapping: a "datanode" is a GPU
(one rank / one device); failover moves a region's device tensors to a
healthy GPU over xGMI (meta/migration.py).
"""

from __future__ import annotations

import time

from greptimedb_amd.meta.failure_detector import PhiAccrualFailureDetector


class RegionSupervisor:
    def __init__(self, on_failover=None, threshold: float = 8.0,
                 acceptable_pause_ms: float = 10_000.0):
        self.detectors: dict[str, PhiAccrualFailureDetector] = {}
        self.on_failover = on_failover          # callback(node_id)
        self.threshold = threshold
        self.acceptable_pause_ms = acceptable_pause_ms
        self.failed: set[str] = set()
        self.maintenance = False   # True = suppress auto-failover
                                   # (reference: metasrv maintenance mode)

    def heartbeat(self, node_id: str, now_ms: float | None = None):
        now_ms = now_ms if now_ms is not None else time.time() * 1000
        det = self.detectors.get(node_id)
        if det is None:
            det = self.detectors[node_id] = PhiAccrualFailureDetector(
                threshold=self.threshold,
                acceptable_heartbeat_pause_ms=self.acceptable_pause_ms)
        det.heartbeat(now_ms)
        self.failed.discard(node_id)

    def check(self, now_ms: float | None = None) -> list[str]:
        """Return nodes whose φ exceeds the threshold; fire failover once.
        In maintenance mode detection still runs but no failover fires
        (planned restarts must not trigger region migration storms)."""
        now_ms = now_ms if now_ms is not None else time.time() * 1000
        if self.maintenance:
            return []
        newly = []
        for node, det in self.detectors.items():
            if node in self.failed:
                continue
            if not det.is_available(now_ms):
                self.failed.add(node)
                newly.append(node)
                if self.on_failover:
                    self.on_failover(node)
        return newly
