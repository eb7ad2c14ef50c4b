"""Step stall diagnostics (the reference had no failure
detection at all — a dead rank hangs every peer's collective until the
30-minute default timeout with zero diagnostics).

This framework already fails collective hangs fast via the configurable
process-group timeout (`core/dist.py`, default 600 s). The watchdog adds the
missing WHY: `faulthandler.dump_traceback_later` is armed at the start of
every training step and cancelled at its end — pure stdlib, one syscall per
step, zero steady-state overhead. If a step exceeds the threshold (a peer
died mid-collective, a deadlocked loader worker, a wedged kernel), every
thread's Python stack is dumped to stderr at the moment of the stall, so
the log shows WHICH collective or load the rank was stuck in rather than a
bare NCCL timeout.
"""
from __future__ import annotations

import faulthandler
from typing import Optional


class StepWatchdog:
    """Arm per step; disarm on completion. No-op when timeout_s is None."""

    def __init__(self, timeout_s: Optional[float], repeat: bool = False):
        self.timeout_s = timeout_s
        self.repeat = repeat
        self._armed = False

    def arm(self) -> None:
        if self.timeout_s is None:
            return
        faulthandler.dump_traceback_later(self.timeout_s, repeat=self.repeat,
                                          exit=False)
        self._armed = True

    def disarm(self) -> None:
        if self._armed:
            faulthandler.cancel_dump_traceback_later()
            self._armed = False

    def __enter__(self):
        self.arm()
        return self

    def __exit__(self, *exc):
        self.disarm()
        return False
