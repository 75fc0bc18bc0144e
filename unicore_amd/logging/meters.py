"""Running-statistic containers backing the metrics aggregator.

Capability parity with the reference meters (unicore/logging/meters.py:
AverageMeter:68, TimeMeter:113, StopwatchMeter:166, MetersDict:222). The
serialized state layouts are kept identical so checkpoints interoperate:
meters are stored by class name + state dict inside ``extra_state.metrics``.

Design differences from the reference: serialization is table-driven off a
per-class ``_state_attrs`` tuple in the base class (one implementation
instead of four), and rounding is a base-class hook.
"""

import time
from collections import OrderedDict
from typing import Dict, Optional

try:
    import torch
except ImportError:  # pragma: no cover - torch is always present in practice
    torch = None

try:
    import numpy as _np
except ImportError:  # pragma: no cover
    _np = None


def safe_round(number, ndigits):
    """Round plain numbers, 0-d tensors and 0-d numpy scalars alike."""
    if hasattr(number, "__round__"):
        return round(number, ndigits)
    if torch is not None and torch.is_tensor(number) and number.numel() == 1:
        return safe_round(number.item(), ndigits)
    if _np is not None and _np.ndim(number) == 0 and hasattr(number, "item"):
        return safe_round(number.item(), ndigits)
    return number


def _match_device(accum, incoming):
    """When accumulating tensors, keep the accumulator on the incoming
    tensor's device/dtype; plain numbers pass through untouched."""
    if torch is None or not torch.is_tensor(accum) or not torch.is_tensor(incoming):
        return accum
    return accum.to(incoming)


class Meter:
    """Common interface: reset, update-style mutation, smoothed readout.

    ``_state_attrs`` names the attributes that round-trip through
    state_dict(); subclasses with extra serialization rules override the
    load path only.
    """

    _state_attrs = ()

    def state_dict(self) -> dict:
        return {name: getattr(self, name) for name in self._state_attrs}

    def load_state_dict(self, state: dict) -> None:
        for name in self._state_attrs:
            if name == "round":
                self.round = state.get("round", None)
            else:
                setattr(self, name, state[name])

    def reset(self) -> None:
        raise NotImplementedError

    def _readout(self):
        raise NotImplementedError

    @property
    def smoothed_value(self):
        """Smoothed scalar for logging (rounded when configured)."""
        value = self._readout()
        ndigits = getattr(self, "round", None)
        if ndigits is not None and value is not None:
            value = safe_round(value, ndigits)
        return value


class AverageMeter(Meter):
    """Weighted running mean; also remembers the latest raw value."""

    _state_attrs = ("val", "sum", "count", "round")

    def __init__(self, round: Optional[int] = None):  # noqa: A002 - API name
        self.round = round
        self.reset()

    def reset(self) -> None:
        self.val = None   # latest update, reported before any weight arrives
        self.sum = 0      # weighted sum of updates
        self.count = 0    # accumulated weight

    def update(self, val, n=1) -> None:
        if val is None:
            return
        self.val = val
        if n > 0:
            self.sum = _match_device(self.sum, val) + val * n
            self.count = _match_device(self.count, n) + n

    @property
    def avg(self):
        return self.sum / self.count if self.count > 0 else self.val

    _readout = avg.fget


class TimeMeter(Meter):
    """Events per second of wall-clock time."""

    _state_attrs = ("init", "n", "round")

    def __init__(self, init: int = 0, n: int = 0,
                 round: Optional[int] = None):  # noqa: A002 - API name
        self.round = round
        self.reset(init, n)

    def reset(self, init=0, n=0) -> None:
        self.init = init              # elapsed time carried over a resume
        self.start = time.perf_counter()
        self.n = n                    # event count
        self.i = 0                    # update() call count

    def update(self, val=1) -> None:
        self.n = _match_device(self.n, val) + val
        self.i += 1

    @property
    def elapsed_time(self):
        return self.init + (time.perf_counter() - self.start)

    @property
    def avg(self):
        return self.n / self.elapsed_time

    _readout = avg.fget

    def state_dict(self) -> dict:
        # freeze the running clock into init so a restore continues from it
        return {"init": self.elapsed_time, "n": self.n, "round": self.round}

    def load_state_dict(self, state: dict) -> None:
        if "start" in state:
            # legacy layout serialized a raw perf_counter origin, which is
            # meaningless across processes; only init survives
            self.reset(init=state["init"])
        else:
            self.reset(init=state["init"], n=state["n"])
            self.round = state.get("round", None)


class StopwatchMeter(Meter):
    """Accumulates durations between explicit start()/stop() calls."""

    _state_attrs = ("sum", "n", "round")

    def __init__(self, round: Optional[int] = None):  # noqa: A002 - API name
        self.round = round
        self.sum = 0
        self.n = 0
        self.start_time = None

    def start(self) -> None:
        self.start_time = time.perf_counter()

    def stop(self, n=1, prehook=None) -> None:
        if self.start_time is None:
            return  # never started: nothing to accumulate
        if prehook is not None:
            prehook()
        self.sum = self.sum + (time.perf_counter() - self.start_time)
        self.n = _match_device(self.n, n) + n

    def reset(self) -> None:
        self.sum = 0
        self.n = 0
        self.start()

    @property
    def avg(self):
        return self.sum / self.n if self.n > 0 else self.sum

    @property
    def elapsed_time(self):
        if self.start_time is None:
            return 0.0
        return time.perf_counter() - self.start_time

    def load_state_dict(self, state: dict) -> None:
        super().load_state_dict(state)
        self.start_time = None

    def _readout(self):
        # while running with nothing accumulated, report the live split
        return self.avg if self.sum > 0 else self.elapsed_time


class MetersDict(OrderedDict):
    """Ordered meter collection, iterated lowest-priority-first.

    Each key is registered once with a fixed priority; ties keep insertion
    order. Keys beginning with ``_`` are internal and excluded from the
    smoothed-value snapshot.
    """

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.priorities = []

    def __setitem__(self, key, value) -> None:
        if key in self:
            raise AssertionError("MetersDict doesn't support reassignment")
        priority, meter = value
        entry = (priority, len(self.priorities), key)
        # insert into the sorted registry, then rebuild the dict ordering
        pos = 0
        while pos < len(self.priorities) and self.priorities[pos] < entry:
            pos += 1
        self.priorities.insert(pos, entry)
        super().__setitem__(key, meter)
        for _, _, k in self.priorities:
            self.move_to_end(k)

    def add_meter(self, key, meter, priority) -> None:
        self[key] = (priority, meter)

    def get_smoothed_value(self, key: str):
        meter = self[key]
        if isinstance(meter, MetersDict._DerivedMeter):
            return meter.fn(self)
        return meter.smoothed_value

    def get_smoothed_values(self) -> Dict[str, float]:
        return OrderedDict(
            (key, self.get_smoothed_value(key))
            for key in self
            if not key.startswith("_")
        )

    def reset(self) -> None:
        for meter in self.values():
            if not isinstance(meter, MetersDict._DerivedMeter):
                meter.reset()

    def state_dict(self) -> list:
        out = []
        for priority, _, key in self.priorities:
            meter = self[key]
            if isinstance(meter, MetersDict._DerivedMeter):
                continue  # derived meters hold a callable; not serializable
            out.append((priority, key, type(meter).__name__, meter.state_dict()))
        return out

    def load_state_dict(self, state: list) -> None:
        self.clear()
        self.priorities.clear()
        for priority, key, cls_name, meter_state in state:
            meter = globals()[cls_name]()
            meter.load_state_dict(meter_state)
            self.add_meter(key, meter, priority)

    class _DerivedMeter(Meter):
        """Read-only meter computed from the other meters at readout time."""

        def __init__(self, fn):
            self.fn = fn

        def reset(self) -> None:
            pass
