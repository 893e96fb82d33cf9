"""Process-global, thread-safe monitoring wrapper (parity: reference monitoring.py).

Wraps ``pipeedge_amd.monitoring.MonitorContext`` in a module-global guarded by
a read/write lock, with per-thread iteration contexts so overlapping threads
can share a key (reference monitoring.py:152-210), periodic window logging,
and the ``CSV_FILE_MODE`` env override (reference monitoring.py:19-20).
"""
import logging
import os
import threading
from contextlib import contextmanager
from typing import Optional, Union

from pipeedge_amd.monitoring import MonitorContext, MonitorIterationContext
from pipeedge_amd.utils.threads import RWLock

logger = logging.getLogger(__name__)

ENV_CSV_FILE_MODE = "CSV_FILE_MODE"
_CSV_FILE_MODE = "w"

PRINT_FIELDS_INSTANT = True
PRINT_FIELDS_WINDOW = True
PRINT_FIELDS_GLOBAL = True

_monitor_ctx: Optional[MonitorContext] = None
_monitor_ctx_lock = RWLock()
_locks = {}
_work_types = {}
_acc_types = {}
_thr_ctx = {}  # (key, thread-ident) -> MonitorIterationContext


def init(key: str, window_size: int, work_type: str = "items",
         acc_type: str = "acc") -> None:
    """Create the global monitoring context."""
    global _monitor_ctx  # noqa: PLW0603
    log_mode = os.getenv(ENV_CSV_FILE_MODE, _CSV_FILE_MODE)
    with _monitor_ctx_lock.lock_write():
        _monitor_ctx = MonitorContext(key=key, window_size=window_size,
                                      log_name=key.replace("/", "_") + ".csv",
                                      log_mode=log_mode)
        logger.info("Monitoring energy source: %s",
                    _monitor_ctx.energy_source.name)
        _monitor_ctx.open()
        _locks[key] = threading.Lock()
        _work_types[key] = work_type
        _acc_types[key] = acc_type


def finish() -> None:
    """Destroy the global monitoring context (logs global stats first)."""
    global _monitor_ctx  # noqa: PLW0603
    with _monitor_ctx_lock.lock_write():
        if _monitor_ctx is None:
            return
        if PRINT_FIELDS_GLOBAL:
            for key in _monitor_ctx.keys():
                _log_global(key)
        _monitor_ctx.close()
        _monitor_ctx = None
        _thr_ctx.clear()
        _locks.clear()
        _work_types.clear()
        _acc_types.clear()


def add_key(key: str, work_type: str = "items", acc_type: str = "acc") -> None:
    with _monitor_ctx_lock.lock_write():
        if _monitor_ctx is None:
            return
        _monitor_ctx.add_heartbeat(key=key,
                                   log_name=key.replace("/", "_") + ".csv")
        _locks[key] = threading.Lock()
        _work_types[key] = work_type
        _acc_types[key] = acc_type


@contextmanager
def get_locked_context(key: str):
    """Yields the MonitorContext with a lock on ``key``."""
    with _monitor_ctx_lock.lock_read():
        with _locks[key]:
            yield _monitor_ctx


def _log_instant(key):
    mctx = _monitor_ctx
    logger.info("%s: instant: heart rate: %.3f /s, perf: %.3f %s/s, acc: %.3f %s",
                key, mctx.get_instant_heart_rate(key), mctx.get_instant_perf(key),
                _work_types.get(key, "items"), mctx.get_instant_accuracy(key),
                _acc_types.get(key, "acc"))


def _log_window(key):
    mctx = _monitor_ctx
    logger.info("%s: window: heart rate: %.3f /s, perf: %.3f %s/s, power: %.3f W",
                key, mctx.get_window_heart_rate(key), mctx.get_window_perf(key),
                _work_types.get(key, "items"), mctx.get_window_power_w(key))


def _log_global(key):
    mctx = _monitor_ctx
    logger.info("%s: global: count: %d, work: %.1f %s, perf: %.3f %s/s, "
                "energy: %.3f J, acc rate: %.3f %s/s",
                key, mctx.get_count(key), mctx.get_global_work(key),
                _work_types.get(key, "items"), mctx.get_global_perf(key),
                _work_types.get(key, "items"), mctx.get_global_energy_j(key),
                mctx.get_global_accuracy_rate(key), _acc_types.get(key, "acc"))


def iteration_start(key: str) -> None:
    """Start an iteration for (key, current thread)."""
    with _monitor_ctx_lock.lock_read():
        if _monitor_ctx is None:
            return
        ident = (key, threading.get_ident())
        with _locks[key]:
            ctx = _thr_ctx.setdefault(ident, MonitorIterationContext())
            _monitor_ctx.iteration_start(key, ctx=ctx)


def iteration(key: str, work: Union[int, float] = 1,
              accuracy: Union[int, float] = 0, safe: bool = True) -> None:
    """Complete an iteration for (key, current thread)."""
    with _monitor_ctx_lock.lock_read():
        if _monitor_ctx is None:
            return
        ident = (key, threading.get_ident())
        with _locks[key]:
            ctx = _thr_ctx.get(ident) if safe else None
            if safe and ctx is None:
                ctx = _thr_ctx.setdefault(ident, MonitorIterationContext())
            _monitor_ctx.iteration(key, work=work, accuracy=accuracy, ctx=ctx)
            count = _monitor_ctx.get_count(key)
            if PRINT_FIELDS_INSTANT:
                _log_instant(key)
            if PRINT_FIELDS_WINDOW and count > 0 and \
                    count % _monitor_ctx.window_size == 0:
                _log_window(key)
