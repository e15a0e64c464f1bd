"""Lightweight per-step GPU timing (SURVEY.md §5.1 tracing/profiling).

HIP events bracket the phases of a training step (forward / backward /
optimizer / comm); results land in the JSONL metrics stream so rocprof
isn't needed for coarse step anatomy.
"""

from __future__ import annotations

import torch


class PhaseTimer:
    """Event-based phase timer. Usage:
        tm = PhaseTimer(enabled=device.type == 'cuda')
        tm.mark('fwd'); ...forward...; tm.mark('bwd'); ...; tm.mark(None)
        tm.collect() -> {'fwd': ms, 'bwd': ms}  (synchronises)
    """

    def __init__(self, enabled: bool = True):
        self.enabled = enabled and torch.cuda.is_available()
        self._events: list[tuple[str | None, torch.cuda.Event]] = []

    def mark(self, phase: str | None):
        if not self.enabled:
            return
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        self._events.append((phase, ev))

    def collect(self) -> dict[str, float]:
        if not self.enabled or len(self._events) < 2:
            self._events.clear()
            return {}
        self._events[-1][1].synchronize()
        out: dict[str, float] = {}
        for (name, ev0), (_, ev1) in zip(self._events, self._events[1:]):
            if name is None:
                continue
            out[name] = out.get(name, 0.0) + ev0.elapsed_time(ev1)
        self._events.clear()
        return out
