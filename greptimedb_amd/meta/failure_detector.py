"""φ-accrual failure detector.

Reference parity: src/meta-srv/src/failure_detector.rs:31-180 (itself the
Hayashibara φ-accrual detector used by Akka/Cassandra): heartbeat inter-
arrival intervals feed a sliding-window normal model; φ(t_now) =
-log10(P(no heartbeat by now)). Same defaults as the reference
(threshold 8, min_std_deviation 100ms, acceptable_heartbeat_pause 10s,
first_heartbeat_estimate 1s).
"""

from __future__ import annotations

import math
from collections import deque


class PhiAccrualFailureDetector:
    def __init__(self, threshold: float = 8.0, max_sample_size: int = 1000,
                 min_std_deviation_ms: float = 100.0,
                 acceptable_heartbeat_pause_ms: float = 10_000.0,
                 first_heartbeat_estimate_ms: float = 1_000.0):
        self.threshold = threshold
        self.min_std = min_std_deviation_ms
        self.acceptable_pause = acceptable_heartbeat_pause_ms
        self.first_estimate = first_heartbeat_estimate_ms
        self.intervals: deque[float] = deque(maxlen=max_sample_size)
        self.last_heartbeat_ms: float | None = None

    def heartbeat(self, now_ms: float):
        if self.last_heartbeat_ms is not None:
            self.intervals.append(now_ms - self.last_heartbeat_ms)
        else:
            # seed like the reference: mean = first_estimate, std = mean/4
            mean = self.first_estimate
            self.intervals.append(mean - mean / 4)
            self.intervals.append(mean + mean / 4)
        self.last_heartbeat_ms = now_ms

    def phi(self, now_ms: float) -> float:
        if self.last_heartbeat_ms is None or not self.intervals:
            return 0.0
        elapsed = now_ms - self.last_heartbeat_ms
        mean = sum(self.intervals) / len(self.intervals)
        var = sum((x - mean) ** 2 for x in self.intervals) / len(self.intervals)
        std = max(math.sqrt(var), self.min_std)
        mean += self.acceptable_pause
        y = (elapsed - mean) / std
        # φ = -log10(e^-t / (1 + e^-t)) computed in a numerically stable
        # form (the reference's logistic approximation of the normal CDF)
        t = y * (1.5976 + 0.070566 * y * y)
        x = -t
        log1pe = x if x > 35 else math.log1p(math.exp(x))
        return max((t + log1pe) / math.log(10), 0.0)

    def is_available(self, now_ms: float) -> bool:
        return self.phi(now_ms) < self.threshold
