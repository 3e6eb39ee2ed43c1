"""EventStorage metric store + writers.

Reference behavior: libai/utils/events.py:69-450 and history_buffer.py.
"""

import datetime
import json
import logging
import os
import time
from collections import defaultdict
from contextlib import contextmanager

import torch

__all__ = [
    "EventStorage",
    "EventWriter",
    "JSONWriter",
    "CommonMetricPrinter",
    "TensorboardXWriter",
    "get_event_storage",
]

_CURRENT_STORAGE_STACK = []


def get_event_storage():
    assert _CURRENT_STORAGE_STACK, (
        "get_event_storage() must be called inside a 'with EventStorage(...)' context"
    )
    return _CURRENT_STORAGE_STACK[-1]


class HistoryBuffer:
    """Ring buffer of (value, iteration) with windowed statistics."""

    def __init__(self, max_length=1000000):
        self._max_length = max_length
        self._data = []
        self._count = 0
        self._global_avg = 0.0

    def update(self, value, iteration=None):
        if iteration is None:
            iteration = self._count
        if len(self._data) == self._max_length:
            self._data.pop(0)
        self._data.append((value, iteration))
        self._count += 1
        self._global_avg += (value - self._global_avg) / self._count

    def latest(self):
        return self._data[-1][0]

    def median(self, window_size):
        return float(
            torch.tensor([x[0] for x in self._data[-window_size:]]).median()
        )

    def avg(self, window_size):
        return float(
            torch.tensor([x[0] for x in self._data[-window_size:]]).mean()
        )

    def global_avg(self):
        return self._global_avg

    def values(self):
        return self._data


class EventStorage:
    def __init__(self, start_iter=0):
        self._history = defaultdict(HistoryBuffer)
        self._smoothing_hints = {}
        self._latest_scalars = {}
        self._iter = start_iter
        self._name_prefix = ""

    @property
    def iter(self):
        return self._iter

    @iter.setter
    def iter(self, val):
        self._iter = int(val)

    def step(self):
        self._iter += 1

    def put_scalar(self, name, value, smoothing_hint=True):
        name = self._name_prefix + name
        value = float(value)
        self._history[name].update(value, self._iter)
        self._latest_scalars[name] = (value, self._iter)
        existing = self._smoothing_hints.get(name)
        if existing is not None:
            assert existing == smoothing_hint, f"inconsistent smoothing for {name}"
        else:
            self._smoothing_hints[name] = smoothing_hint

    def put_scalars(self, *, smoothing_hint=True, **kwargs):
        for k, v in kwargs.items():
            self.put_scalar(k, v, smoothing_hint=smoothing_hint)

    def history(self, name):
        ret = self._history.get(name)
        if ret is None:
            raise KeyError(f"no history metric {name!r}")
        return ret

    def histories(self):
        return self._history

    def latest(self):
        return self._latest_scalars

    def latest_with_smoothing_hint(self, window_size=20):
        result = {}
        for k, (v, itr) in self._latest_scalars.items():
            result[k] = (
                self._history[k].median(window_size)
                if self._smoothing_hints.get(k)
                else v,
                itr,
            )
        return result

    def smoothing_hints(self):
        return self._smoothing_hints

    @contextmanager
    def name_scope(self, name):
        old = self._name_prefix
        self._name_prefix = name.rstrip("/") + "/"
        yield
        self._name_prefix = old

    def clear_metrics(self):
        self._latest_scalars = {}

    def __enter__(self):
        _CURRENT_STORAGE_STACK.append(self)
        return self

    def __exit__(self, *args):
        assert _CURRENT_STORAGE_STACK[-1] is self
        _CURRENT_STORAGE_STACK.pop()


class EventWriter:
    def write(self):
        raise NotImplementedError

    def close(self):
        pass


class JSONWriter(EventWriter):
    """Append latest scalars as JSON lines (reference: events.py:69-135)."""

    def __init__(self, json_file, window_size=20):
        os.makedirs(os.path.dirname(json_file) or ".", exist_ok=True)
        self._file = open(json_file, "a")
        self._window_size = window_size
        self._last_write = -1

    def write(self):
        storage = get_event_storage()
        to_save = defaultdict(dict)
        for k, (v, itr) in storage.latest_with_smoothing_hint(self._window_size).items():
            if itr <= self._last_write:
                continue
            to_save[itr][k] = v
        if to_save:
            self._last_write = max(to_save.keys())
        for itr, scalars in sorted(to_save.items()):
            scalars["iteration"] = itr
            self._file.write(json.dumps(scalars, sort_keys=True) + "\n")
        self._file.flush()

    def close(self):
        self._file.close()


class TensorboardXWriter(EventWriter):
    def __init__(self, log_dir, window_size=20):
        self._window_size = window_size
        self._writer = None
        try:
            from torch.utils.tensorboard import SummaryWriter

            self._writer = SummaryWriter(log_dir)
        except Exception:
            logging.getLogger(__name__).warning(
                "tensorboard unavailable; TensorboardXWriter disabled"
            )
        self._last_write = -1

    def write(self):
        if self._writer is None:
            return
        storage = get_event_storage()
        new_last = self._last_write
        for k, (v, itr) in storage.latest_with_smoothing_hint(self._window_size).items():
            if itr > self._last_write:
                self._writer.add_scalar(k, v, itr)
                new_last = max(new_last, itr)
        self._last_write = new_last

    def close(self):
        if self._writer:
            self._writer.close()


class CommonMetricPrinter(EventWriter):
    """Console writer with ETA + throughput (reference: events.py:178-263)."""

    def __init__(self, batch_size, max_iter, window_size=20):
        self.logger = logging.getLogger("libai_amd.utils.events")
        self._batch_size = batch_size
        self._max_iter = max_iter
        self._window_size = window_size

    def write(self):
        storage = get_event_storage()
        iteration = storage.iter
        if iteration >= self._max_iter:
            return
        try:
            t = storage.history("time").median(self._window_size)
            eta = str(datetime.timedelta(seconds=int(t * (self._max_iter - iteration))))
            tput = self._batch_size / t
            time_str = f"time: {t:.4f} s/iter  throughput: {tput:.1f} samples/s  eta: {eta}"
        except KeyError:
            time_str = ""
        losses = "  ".join(
            f"{k}: {v:.4g}"
            for k, (v, _) in storage.latest_with_smoothing_hint(self._window_size).items()
            if "loss" in k
        )
        lr = ""
        try:
            lr = f"lr: {storage.history('lr').latest():.3e}"
        except KeyError:
            pass
        mem = ""
        if torch.cuda.is_available():
            mem = f"max_mem: {torch.cuda.max_memory_allocated() / 2**20:.0f}M"
        self.logger.info(
            f"iter: {iteration}/{self._max_iter}  {losses}  {time_str}  {lr}  {mem}"
        )
