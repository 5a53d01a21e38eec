"""Profiling utilities (the reference has none — SURVEY.md §5.1).

* ``StepTimer`` — per-step wall breakdown (data / h2d / forward /
  backward / optimizer) with HIP-event timing on GPU so the phases are
  device-accurate, monotonic-clock timing on CPU.
* ``torch_profile`` — context manager wrapping ``torch.profiler`` for N
  steps with a chrome-trace export readable by perfetto / chrome, and
  rocprofv3-friendly (kernels appear under their real gfx950 names).
"""

from __future__ import annotations

import contextlib
import os
import time
from collections import defaultdict
from typing import Dict, Iterator, Optional

import torch

__all__ = ["StepTimer", "torch_profile"]


class StepTimer:
    """Accumulates per-phase times across steps.

    Usage::

        timer = StepTimer(enabled=True)
        with timer.phase("data"):   batch = next(it)
        with timer.phase("forward"): loss = model(batch)
        ...
        timer.step_end()
        print(timer.summary())
    """

    PHASES = ("data", "h2d", "forward", "backward", "optimizer", "other")

    def __init__(self, enabled: bool = True, use_cuda: Optional[bool] = None):
        self.enabled = enabled
        self.use_cuda = (
            torch.cuda.is_available() if use_cuda is None else use_cuda
        )
        self.totals: Dict[str, float] = defaultdict(float)
        self.steps = 0
        self._events = []  # (name, start_evt, end_evt) for pending cuda pairs

    @contextlib.contextmanager
    def phase(self, name: str) -> Iterator[None]:
        if not self.enabled:
            yield
            return
        if self.use_cuda:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            yield
            end.record()
            self._events.append((name, start, end))
        else:
            t0 = time.perf_counter()
            yield
            self.totals[name] += time.perf_counter() - t0

    def step_end(self) -> None:
        if not self.enabled:
            return
        if self._events:
            torch.cuda.synchronize()
            for name, start, end in self._events:
                self.totals[name] += start.elapsed_time(end) / 1e3
            self._events.clear()
        self.steps += 1

    def summary(self) -> Dict[str, float]:
        """Mean seconds per step per phase (plus 'total')."""
        if self.steps == 0:
            return {}
        out = {k: v / self.steps for k, v in self.totals.items()}
        out["total"] = sum(v for k, v in out.items())
        return out

    def format_summary(self) -> str:
        s = self.summary()
        if not s:
            return "StepTimer: no steps recorded"
        parts = [
            f"{k}={s[k] * 1e3:.2f}ms" for k in self.PHASES if k in s
        ] + [f"total={s['total'] * 1e3:.2f}ms"]
        return "step breakdown: " + " ".join(parts)


@contextlib.contextmanager
def torch_profile(
    out_dir: str,
    enabled: bool = True,
    with_stack: bool = False,
    rank: int = 0,
) -> Iterator[Optional[torch.profiler.profile]]:
    """Profile the enclosed region; writes chrome trace + op table.

    Produces ``{out_dir}/trace_rank{rank}.json`` (perfetto/chrome) and
    ``{out_dir}/ops_rank{rank}.txt`` (self-time table).
    """
    if not enabled:
        yield None
        return
    os.makedirs(out_dir, exist_ok=True)
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    with torch.profiler.profile(
        activities=activities,
        record_shapes=False,
        with_stack=with_stack,
    ) as prof:
        yield prof
    prof.export_chrome_trace(os.path.join(out_dir, f"trace_rank{rank}.json"))
    table = prof.key_averages().table(
        sort_by=(
            "self_cuda_time_total"
            if torch.cuda.is_available()
            else "self_cpu_time_total"
        ),
        row_limit=60,
    )
    with open(os.path.join(out_dir, f"ops_rank{rank}.txt"), "w") as f:
        f.write(table + "\n")
