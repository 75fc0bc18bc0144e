"""Nested metric aggregation contexts.

Capability parity with the reference aggregator (unicore/logging/metrics.py:
aggregate:45, log_scalar:112, log_speed:149, log_start_time:171,
state_dict:281): every ``log_*`` call fans out to all currently-active
aggregation scopes, each scope is a priority-ordered :class:`MetersDict`,
and the whole registry serializes into checkpoints.

Structurally this implementation keeps the registry state in one
``_Registry`` object (instead of three module globals) and funnels all the
``log_*`` entry points through a single ensure-then-update helper.
"""

import contextlib
import uuid
from collections import defaultdict
from typing import Callable, List, Optional

from .meters import AverageMeter, Meter, MetersDict, StopwatchMeter, TimeMeter


class _Registry:
    """All aggregation state: named scopes plus the active-scope refcounts."""

    def __init__(self):
        self.scopes = {}
        self.active = {}
        self.refcount = defaultdict(int)
        self.clear()

    def clear(self):
        self.scopes.clear()
        self.active.clear()
        self.refcount.clear()
        # the default scope observes everything logged anywhere
        root = MetersDict()
        self.scopes["default"] = root
        self.active["default"] = root
        self.refcount["default"] = 1


_R = _Registry()


def reset() -> None:
    """Drop every aggregator and re-create the default scope."""
    _R.clear()


@contextlib.contextmanager
def aggregate(name: Optional[str] = None, new_root: bool = False):
    """Collect metrics under *name* for the duration of the block.

    Scopes nest; values logged inside the block land in every active scope.
    With ``new_root`` the block REPLACES the active set instead of joining
    it (validation uses this to keep train meters untouched, reference
    unicore_cli/train.py:377).
    """
    if name is None:
        name = str(uuid.uuid4())  # anonymous one-shot scope
        assert name not in _R.scopes
        scope = MetersDict()
    else:
        assert name != "default"
        scope = _R.scopes.setdefault(name, MetersDict())

    if new_root:
        saved_active, saved_counts = dict(_R.active), dict(_R.refcount)
        _R.active.clear()
        _R.refcount.clear()

    _R.active[name] = scope
    _R.refcount[name] += 1
    yield scope
    _R.refcount[name] -= 1
    if _R.refcount[name] == 0:
        _R.active.pop(name, None)

    if new_root:
        _R.active.clear()
        _R.active.update(saved_active)
        _R.refcount.clear()
        _R.refcount.update(saved_counts)


def get_active_aggregators() -> List[MetersDict]:
    return list(_R.active.values())


def _fanout(key, make_meter, act=None, priority=10, on_create=None):
    """For each active scope: create the meter on first sight, then apply
    *act* (or *on_create* right after creation, for meters that self-start)."""
    for scope in _R.active.values():
        created = key not in scope
        if created:
            scope.add_meter(key, make_meter(), priority)
            if on_create is not None:
                on_create(scope[key])
        if act is not None and (on_create is None or not created):
            act(scope[key])


def log_scalar(key: str, value: float, weight: float = 1, priority: int = 10,
               round: Optional[int] = None) -> None:
    """Weighted-average scalar."""
    _fanout(key, lambda: AverageMeter(round=round),
            act=lambda m: m.update(value, weight), priority=priority)


def log_derived(key: str, fn: Callable[[MetersDict], float],
                priority: int = 20) -> None:
    """Scalar computed from other meters at readout time."""
    _fanout(key, lambda: MetersDict._DerivedMeter(fn), priority=priority)


def log_speed(key: str, value: float, priority: int = 30,
              round: Optional[int] = None) -> None:
    """Events per second; the first call only starts the clock."""
    _fanout(key, lambda: TimeMeter(round=round),
            act=lambda m: m.update(value), priority=priority,
            on_create=lambda m: m.reset())


def log_start_time(key: str, priority: int = 40,
                   round: Optional[int] = None) -> None:
    """Open a stopwatch interval."""
    _fanout(key, lambda: StopwatchMeter(round=round),
            act=lambda m: m.start(), priority=priority)


def log_stop_time(key: str, weight: float = 0.0, prehook=None) -> None:
    """Close a stopwatch interval (no-op where the key never started)."""
    for scope in _R.active.values():
        if key in scope:
            scope[key].stop(weight, prehook)


def log_custom(new_meter_fn: Callable[[], Meter], key: str, *args,
               priority: int = 50, **kwargs) -> None:
    """Route updates into a user-supplied Meter type."""
    _fanout(key, new_meter_fn,
            act=lambda m: m.update(*args, **kwargs), priority=priority)


def reset_meter(name: str, key: str) -> None:
    meter = get_meter(name, key)
    if meter is not None:
        meter.reset()


def reset_meters(name: str) -> None:
    scope = get_meters(name)
    if scope is not None:
        scope.reset()


def get_meter(name: str, key: str) -> Meter:
    scope = _R.scopes.get(name)
    return scope.get(key, None) if scope is not None else None


def get_meters(name: str) -> MetersDict:
    return _R.scopes.get(name, None)


def get_smoothed_value(name: str, key: str) -> float:
    return _R.scopes[name].get_smoothed_value(key)


def get_smoothed_values(name: str):
    return _R.scopes[name].get_smoothed_values()


def state_dict() -> dict:
    return {name: scope.state_dict() for name, scope in _R.scopes.items()}


def load_state_dict(state: dict) -> None:
    for name, scope_state in state.items():
        restored = MetersDict()
        restored.load_state_dict(scope_state)
        _R.scopes[name] = restored
