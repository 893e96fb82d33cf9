"""Keyed heartbeat monitoring core.

Capability parity with reference ``src/pipeedge/monitoring/__init__.py``
(Application-Heartbeats-style keyed metrics with instant / window / global
scopes, CSV records per heartbeat, optional energy counters). The reference
used the ``apphb`` + ``energymon`` native libraries; this build implements the
same record/rate accounting natively and reads energy from AMD GPU hwmon
counters (``energy*_input``, microjoules) when present - the MI355X
equivalent of EnergyMon.
"""
import csv
import glob
import os
import time
from typing import Dict, List, Optional


class EnergySource:
    """Energy metering from hwmon (the EnergyMon equivalent - SURVEY.md 5.5).

    Prefers cumulative ``energy*_input`` counters (microjoules); falls back
    to ``power*_input`` (microwatts, what MI355X boxes expose), which the
    monitor integrates over each iteration (trapezoid)."""

    def __init__(self):
        self._files = []
        self.kind = "none"
        for path in sorted(glob.glob("/sys/class/hwmon/hwmon*/energy*_input")):
            if self._probe(path):
                self._files.append(path)
        if self._files:
            self.kind = "energy"
        else:
            for path in sorted(
                    glob.glob("/sys/class/hwmon/hwmon*/power*_input")):
                if self._probe(path):
                    self._files.append(path)
            if self._files:
                self.kind = "power"
        self.name = "hwmon-" + self.kind if self._files else "none"

    @staticmethod
    def _probe(path) -> bool:
        try:
            with open(path) as f:
                int(f.read().strip())
            return True
        except (OSError, ValueError):
            return False

    def _read_sum(self) -> int:
        total = 0
        for path in self._files:
            try:
                with open(path) as f:
                    total += int(f.read().strip())
            except (OSError, ValueError):
                pass
        return total

    def read_uj(self) -> int:
        """Cumulative energy in microjoules (energy-counter sources only)."""
        return self._read_sum() if self.kind == "energy" else 0

    def read_uw(self) -> int:
        """Instantaneous power in microwatts (power sources only)."""
        return self._read_sum() if self.kind == "power" else 0


class _HeartbeatRecord:
    __slots__ = ("ident", "tag", "time_s", "work", "energy_uj", "accuracy")

    def __init__(self, ident, tag, time_s, work, energy_uj, accuracy):
        self.ident = ident
        self.tag = tag
        self.time_s = time_s
        self.work = work
        self.energy_uj = energy_uj
        self.accuracy = accuracy


class Heartbeat:
    """Per-key iteration accounting with instant/window/global scopes."""

    _CSV_FIELDS = ["id", "tag", "time_s", "work", "energy_uj", "accuracy"]

    def __init__(self, key: str, window_size: int, log_name: Optional[str],
                 log_mode: str = "w"):
        self.key = key
        self.window_size = window_size
        self.records: List[_HeartbeatRecord] = []
        self._log_file = None
        self._log_writer = None
        if log_name:
            self._log_file = open(log_name, log_mode, newline="")
            self._log_writer = csv.writer(self._log_file)
            if log_mode == "w" or self._log_file.tell() == 0:
                self._log_writer.writerow(self._CSV_FIELDS)

    def heartbeat(self, ident: int, tag: int, time_s: float, work: float,
                  energy_uj: int, accuracy: float):
        rec = _HeartbeatRecord(ident, tag, time_s, work, energy_uj, accuracy)
        self.records.append(rec)
        if self._log_writer is not None:
            self._log_writer.writerow([ident, tag, f"{time_s:.6f}", work,
                                       energy_uj, accuracy])
            self._log_file.flush()

    def close(self):
        if self._log_file is not None:
            self._log_file.close()
            self._log_file = None

    # -- scope helpers -------------------------------------------------------
    def _scope(self, scope: str) -> List[_HeartbeatRecord]:
        if scope == "instant":
            return self.records[-1:]
        if scope == "window":
            return self.records[-self.window_size:]
        return self.records

    def count(self) -> int:
        return len(self.records)

    def get_time_s(self, scope="global") -> float:
        return sum(r.time_s for r in self._scope(scope))

    def get_heart_rate(self, scope="global") -> float:
        recs = self._scope(scope)
        t = sum(r.time_s for r in recs)
        return len(recs) / t if t > 0 else 0.0

    def get_work(self, scope="global") -> float:
        return sum(r.work for r in self._scope(scope))

    def get_perf(self, scope="global") -> float:
        recs = self._scope(scope)
        t = sum(r.time_s for r in recs)
        return sum(r.work for r in recs) / t if t > 0 else 0.0

    def get_energy_j(self, scope="global") -> float:
        return sum(r.energy_uj for r in self._scope(scope)) / 1e6

    def get_power_w(self, scope="global") -> float:
        recs = self._scope(scope)
        t = sum(r.time_s for r in recs)
        return (sum(r.energy_uj for r in recs) / 1e6) / t if t > 0 else 0.0

    def get_accuracy(self, scope="global") -> float:
        return sum(r.accuracy for r in self._scope(scope))

    def get_accuracy_rate(self, scope="global") -> float:
        recs = self._scope(scope)
        t = sum(r.time_s for r in recs)
        return sum(r.accuracy for r in recs) / t if t > 0 else 0.0


class MonitorIterationContext:
    """Tracks one in-flight iteration (start time + start energy/power)."""

    def __init__(self):
        self.t_start = None
        self.e_start = 0
        self.p_start = 0

    def start(self, energy_source: Optional[EnergySource]):
        self.t_start = time.perf_counter()
        if energy_source is not None:
            self.e_start = energy_source.read_uj()
            self.p_start = energy_source.read_uw()
        else:
            self.e_start = 0
            self.p_start = 0


class MonitorContext:
    """Multi-key monitor (parity: reference MonitorContext,
    monitoring/__init__.py:61-330)."""

    def __init__(self, key: str, window_size: int = 10,
                 log_name: Optional[str] = None, log_mode: str = "w",
                 use_energy: bool = True):
        self.window_size = window_size
        self._default_key = key
        self.energy_source = EnergySource() if use_energy else None
        self._log_mode = log_mode
        self._hbs: Dict[str, Heartbeat] = {}
        self._tags: Dict[str, int] = {}
        self._iter_ctx: Dict[str, MonitorIterationContext] = {}
        self.add_heartbeat(key, log_name)
        self._open = False

    def open(self):
        self._open = True

    def close(self):
        for hb in self._hbs.values():
            hb.close()
        self._open = False

    def keys(self):
        return list(self._hbs.keys())

    def add_heartbeat(self, key: str, log_name: Optional[str] = None):
        self._hbs[key] = Heartbeat(key, self.window_size, log_name, self._log_mode)
        self._tags[key] = 0

    def iteration_start(self, key: Optional[str] = None,
                        ctx: Optional[MonitorIterationContext] = None):
        key = key or self._default_key
        if ctx is None:
            ctx = self._iter_ctx.setdefault(key, MonitorIterationContext())
        ctx.start(self.energy_source)

    def iteration(self, key: Optional[str] = None, work: float = 1,
                  accuracy: float = 0,
                  ctx: Optional[MonitorIterationContext] = None):
        key = key or self._default_key
        if ctx is None:
            ctx = self._iter_ctx.setdefault(key, MonitorIterationContext())
        now = time.perf_counter()
        t_start = ctx.t_start if ctx.t_start is not None else now
        energy = 0
        if self.energy_source is not None and ctx.t_start is not None:
            if self.energy_source.kind == "energy":
                energy = self.energy_source.read_uj() - ctx.e_start
            elif self.energy_source.kind == "power":
                # trapezoid integration of the power samples: uW * s = uJ
                p_end = self.energy_source.read_uw()
                energy = int(0.5 * (ctx.p_start + p_end) * (now - t_start))
        hb = self._hbs[key]
        tag = self._tags[key]
        self._tags[key] = tag + 1
        hb.heartbeat(tag, tag, now - t_start, work, max(energy, 0), accuracy)
        ctx.t_start = None

    # -- metric getters (all scopes) -----------------------------------------
    def _get(self, key, attr, scope):
        return getattr(self._hbs[key or self._default_key], attr)(scope)

    def get_count(self, key=None):
        return self._hbs[key or self._default_key].count()

    def get_tag(self, key=None):
        """Iteration counter for a key (next tag to be assigned)."""
        return self._tags[key or self._default_key]

    def get_window_size(self, key=None):
        return self.window_size

    def get_window_heartrate(self, key=None):
        return self._hbs[key or self._default_key].get_heart_rate("window")

    def __getattr__(self, name):
        # get_{instant,window,global}_{metric} dynamic getters
        for scope in ("instant", "window", "global"):
            prefix = f"get_{scope}_"
            if name.startswith(prefix):
                attr = "get_" + name[len(prefix):]
                return lambda key=None, a=attr, s=scope: self._get(key, a, s)
        raise AttributeError(name)

    def __enter__(self):
        self.open()
        return self

    def __exit__(self, *args):
        self.close()
