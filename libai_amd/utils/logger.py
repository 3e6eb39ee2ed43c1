"""Rank-aware logging (reference: libai/utils/logger.py:56-214)."""

import atexit
import functools
import logging
import os
import sys
from collections import Counter

__all__ = ["setup_logger", "log_first_n", "log_every_n", "log_every_n_seconds"]


class _ColorFormatter(logging.Formatter):
    GREY, YELLOW, RED, RESET = "\x1b[38m", "\x1b[33m", "\x1b[31m", "\x1b[0m"

    def format(self, record):
        msg = super().format(record)
        if record.levelno == logging.WARNING:
            return self.YELLOW + msg + self.RESET
        if record.levelno >= logging.ERROR:
            return self.RED + msg + self.RESET
        return msg


@functools.lru_cache()
def setup_logger(output=None, distributed_rank=0, *, color=True, name="libai_amd",
                 abbrev_name=None):
    logger = logging.getLogger(name)
    logger.setLevel(logging.DEBUG)
    logger.propagate = False

    fmt = logging.Formatter(
        "[%(asctime)s] %(name)s %(levelname)s: %(message)s", datefmt="%m/%d %H:%M:%S"
    )
    if distributed_rank == 0:
        ch = logging.StreamHandler(stream=sys.stdout)
        ch.setLevel(logging.DEBUG)
        ch.setFormatter(
            _ColorFormatter(
                "[%(asctime)s] %(name)s %(levelname)s: %(message)s",
                datefmt="%m/%d %H:%M:%S",
            )
            if color and sys.stdout.isatty()
            else fmt
        )
        logger.addHandler(ch)

    if output is not None:
        filename = output if output.endswith(".txt") or output.endswith(".log") else os.path.join(
            output, "log.txt"
        )
        if distributed_rank > 0:
            filename = filename + f".rank{distributed_rank}"
        os.makedirs(os.path.dirname(filename) or ".", exist_ok=True)
        fh = logging.StreamHandler(_cached_log_stream(filename))
        fh.setLevel(logging.DEBUG)
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    return logger


@functools.lru_cache(maxsize=None)
def _cached_log_stream(filename):
    io = open(filename, "a", buffering=1024)
    atexit.register(io.close)
    return io


_LOG_COUNTER = Counter()
_LOG_TIMER = {}


def _caller_key():
    frame = sys._getframe(3)
    return (frame.f_code.co_filename, frame.f_lineno)


def log_first_n(lvl, msg, n=1, *, name=None, key="caller"):
    k = _caller_key() if key == "caller" else (key,)
    _LOG_COUNTER[k] += 1
    if _LOG_COUNTER[k] <= n:
        logging.getLogger(name or "libai_amd").log(lvl, msg)


def log_every_n(lvl, msg, n=1, *, name=None):
    k = _caller_key()
    _LOG_COUNTER[k] += 1
    if n == 1 or _LOG_COUNTER[k] % n == 1:
        logging.getLogger(name or "libai_amd").log(lvl, msg)


def log_every_n_seconds(lvl, msg, n=1, *, name=None):
    import time

    k = _caller_key()
    last = _LOG_TIMER.get(k)
    now = time.time()
    if last is None or now - last >= n:
        logging.getLogger(name or "libai_amd").log(lvl, msg)
        _LOG_TIMER[k] = now
