"""Token-bound scale arbiter — the fair-throttle control law.

One multiplier per device, applied by every co-located container's
limiter to its entitled CU share (``RATE_FULL x cu% x scale``).  The
input per tick is (active, bound): how many CU-limited containers are
launching, and how many of those sampled token-bound (bucket <= 0).

Target: HALF the active containers bound.  Heavy containers bind first
as the scale drops, so equilibrium clips them at a common token rate
near the median demand while light containers run free below it —
single-knob max-min fairness from state the monitor actually owns (the
buckets in the regions).  Host busy% is deliberately NOT used: on
multi-DRM hosts the wrong card reads ~0 busy and an AIMD-on-busy
controller saturates the scale, unthrottling everyone (measured 19x /
2.1x spreads; profiles/r02_summary.md — the bucket controller measured
1.002x).

Dynamics: RATE_FULL is a static calibration of chip workgroup
throughput, and real workloads sit orders of magnitude off it (measured:
conv kernels retire ~80x more workgroups/s than the constant assumes).
A fixed gentle step would need minutes to traverse that range, so the
arbiter runs SLOW-START: coarse multiplicative moves (x1.25 / x0.8)
until the bound-fraction first crosses the target, then a gentle band
(x1.02 / x0.97) for low-amplitude sawtooth at the fairness edge.  If the
input stays saturated in one direction for ``COARSE_REENTRY`` ticks
(workload change), coarse mode re-arms.
"""
from __future__ import annotations

from dataclasses import dataclass

SCALE_LO, SCALE_HI = 0.05, 1000.0
COARSE_UP, COARSE_DOWN = 1.25, 0.80
FINE_UP, FINE_DOWN = 1.02, 0.97
COARSE_REENTRY = 16  # ticks of saturated input before re-entering coarse


@dataclass
class ScaleArbiter:
    scale: float = 1.0
    coarse: bool = True
    _last_dir: int = 0       # +1 relax, -1 tighten
    _saturated_ticks: int = 0

    def tick(self, active: int, bound: int) -> float:
        """Feed one observation; returns the scale to publish."""
        if active <= 0:
            return self.scale
        relax = 2 * bound >= active
        direction = 1 if relax else -1

        # saturation bookkeeping: all-bound or none-bound streaks mean the
        # operating point is far away — re-arm coarse mode
        saturated = bound >= active or bound == 0
        if saturated and direction == self._last_dir:
            self._saturated_ticks += 1
            if self._saturated_ticks >= COARSE_REENTRY:
                self.coarse = True
        else:
            self._saturated_ticks = 0

        if self.coarse and self._last_dir and direction != self._last_dir:
            self.coarse = False  # crossed the edge: switch to fine band
        self._last_dir = direction

        if self.coarse:
            step = COARSE_UP if relax else COARSE_DOWN
        else:
            step = FINE_UP if relax else FINE_DOWN
        self.scale = max(SCALE_LO, min(SCALE_HI, self.scale * step))
        return self.scale
