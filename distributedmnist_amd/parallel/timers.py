"""Per-step timing: hipEvent pairs on GPU, wall-clock on CPU.

Replaces the reference's Twisted-RPC worker-time instrumentation
(timeout_manager.py:48-70): each rank brackets its local compute with
hipEvents; the sync engine all-gathers the values for the CDF report.
"""

from __future__ import annotations

import time

import torch


class StepTimer:
    def __init__(self, device):
        self.device = torch.device(device)
        self.use_events = self.device.type == "cuda"
        if self.use_events:
            self._e0 = torch.cuda.Event(enable_timing=True)
            self._e1 = torch.cuda.Event(enable_timing=True)
        self._t0 = 0.0

    def start(self):
        if self.use_events:
            self._e0.record()
        self._t0 = time.time()

    def stop(self) -> float:
        """Returns elapsed compute seconds (synchronizes on GPU)."""
        if self.use_events:
            self._e1.record()
            self._e1.synchronize()
            return self._e0.elapsed_time(self._e1) / 1000.0
        return time.time() - self._t0
