"""Pausable wall timer (reference: libai/utils/timer.py)."""

import time

__all__ = ["Timer"]


class Timer:
    def __init__(self):
        self.reset()

    def reset(self):
        self._start = time.perf_counter()
        self._paused = None
        self._total_paused = 0.0
        self._count_start = 1

    def pause(self):
        if self._paused is not None:
            raise ValueError("timer already paused")
        self._paused = time.perf_counter()

    def is_paused(self):
        return self._paused is not None

    def resume(self):
        if self._paused is None:
            raise ValueError("timer not paused")
        self._total_paused += time.perf_counter() - self._paused
        self._paused = None
        self._count_start += 1

    def seconds(self):
        end = self._paused if self._paused is not None else time.perf_counter()
        return end - self._start - self._total_paused

    def avg_seconds(self):
        return self.seconds() / self._count_start
