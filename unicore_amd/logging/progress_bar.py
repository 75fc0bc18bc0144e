"""Progress reporting for training/validation iterators.

Capability parity with the reference bars (unicore/logging/progress_bar.py:
json:138, noop:189, simple:208, tqdm:243, tensorboard wrapper:302,
wandb:313-327): a ``progress_bar()`` factory returns an iterable wrapper with
``log()`` (periodic, mid-epoch) and ``print()`` (end-of-epoch) hooks; the
tensorboard/wandb sink wraps any of the text bars and is only built on the
rank that asks for it.

Structure here differs from the reference: the position-tracking iteration
shared by the json and simple bars lives in one ``_CountingBar`` mixin, and
the format name -> class mapping is a registry dict.
"""

import json
import logging
import os
import sys
from collections import OrderedDict
from contextlib import contextmanager
from numbers import Number
from typing import Optional

import torch

from .meters import AverageMeter, StopwatchMeter, TimeMeter

logger = logging.getLogger(__name__)


@contextmanager
def rename_logger(logger, new_name):
    """Temporarily relabel a logger (used to tag train/valid lines)."""
    saved = logger.name
    if new_name is not None:
        logger.name = new_name
    yield logger
    logger.name = saved


def format_stat(stat):
    """Render one stat value (number / meter / tensor) for display."""
    if isinstance(stat, Number):
        return f"{stat:g}"
    if isinstance(stat, AverageMeter):
        return f"{stat.avg:.3f}"
    if isinstance(stat, TimeMeter):
        return f"{round(stat.avg):g}"
    if isinstance(stat, StopwatchMeter):
        return f"{round(stat.sum):g}"
    if torch.is_tensor(stat):
        return stat.tolist()
    return stat


class BaseProgressBar:
    """Iterable wrapper with log/print hooks; subclasses pick the sink."""

    def __init__(self, iterable, epoch=None, prefix=None):  # noqa: D107
        self.iterable = iterable
        self.n = getattr(iterable, "n", 0)  # resume offset within the epoch
        self.epoch = epoch
        parts = []
        if epoch is not None:
            parts.append(f"epoch {epoch:03d}")
        if prefix is not None:
            parts.append(prefix)
        self.prefix = " | ".join(parts)

    def __len__(self):
        return len(self.iterable)

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        return False

    def __iter__(self):  # subclasses drive the wrapped iterable
        raise NotImplementedError

    def log(self, stats, tag=None, step=None) -> None:
        """Periodic mid-epoch stats."""
        raise NotImplementedError

    def print(self, stats, tag=None, step=None) -> None:
        """End-of-epoch stats."""
        raise NotImplementedError

    def _format_stats(self, stats):
        return OrderedDict((k, str(format_stat(v))) for k, v in stats.items())

    @staticmethod
    def _str_commas(stats):
        return ", ".join(f"{k}={v.strip()}" for k, v in stats.items())

    @staticmethod
    def _str_pipes(stats):
        return " | ".join(f"{k} {v.strip()}" for k, v in stats.items())


class _CountingBar(BaseProgressBar):
    """Shared iteration bookkeeping: tracks position + total and rate-limits
    log() to every ``log_interval`` steps."""

    def __init__(self, iterable, epoch=None, prefix=None, log_interval=1000):
        super().__init__(iterable, epoch=epoch, prefix=prefix)
        self.log_interval = log_interval
        self.i = self.size = None

    def __iter__(self):
        self.size = len(self.iterable)
        position = self.n
        for obj in self.iterable:
            self.i = position
            position += 1
            yield obj

    def _should_log(self, step):
        step = (step if step else self.i) or 0
        return (
            step > 0
            and self.log_interval is not None
            and step % self.log_interval == 0
        )


class JsonProgressBar(_CountingBar):
    """One JSON object per log line (machine-readable)."""

    def log(self, stats, tag=None, step=None) -> None:
        if not self._should_log(step):
            return
        update = None
        if self.epoch is not None:
            update = self.epoch - 1 + (self.i + 1) / float(self.size)
        payload = self._jsonify(stats, epoch=self.epoch, update=update)
        with rename_logger(logger, tag) as sink:
            sink.info(json.dumps(payload))

    def print(self, stats, tag=None, step=None) -> None:
        if tag is not None:
            stats = OrderedDict((f"{tag}_{k}", v) for k, v in stats.items())
        self.stats = stats
        payload = self._jsonify(stats, epoch=self.epoch)
        with rename_logger(logger, tag) as sink:
            sink.info(json.dumps(payload))

    @staticmethod
    def _jsonify(stats, epoch=None, update=None):
        out = OrderedDict()
        if epoch is not None:
            out["epoch"] = epoch
        if update is not None:
            out["update"] = round(update, 3)
        for k, v in stats.items():
            out[k] = format_stat(v)
        return out


class NoopProgressBar(BaseProgressBar):
    """Swallows everything (non-master ranks)."""

    def __iter__(self):
        return iter(self.iterable)

    def log(self, stats, tag=None, step=None) -> None:
        pass

    def print(self, stats, tag=None, step=None) -> None:
        pass


class SimpleProgressBar(_CountingBar):
    """Plain log lines for non-TTY environments."""

    def log(self, stats, tag=None, step=None) -> None:
        if not self._should_log(step):
            return
        body = self._str_commas(self._format_stats(stats))
        with rename_logger(logger, tag) as sink:
            sink.info(f"{self.prefix}:  {self.i + 1:5d} / {self.size:d} {body}")

    def print(self, stats, tag=None, step=None) -> None:
        body = self._str_pipes(self._format_stats(stats))
        with rename_logger(logger, tag) as sink:
            sink.info(f"{self.prefix} | {body}")


class TqdmProgressBar(BaseProgressBar):
    """Interactive tqdm bar with a postfix stats line."""

    def __init__(self, iterable, epoch=None, prefix=None):
        super().__init__(iterable, epoch=epoch, prefix=prefix)
        try:
            from tqdm import tqdm
        except ImportError:
            # degrade gracefully where tqdm is unavailable
            self.tqdm = None
            self._fallback = SimpleProgressBar(iterable, epoch, prefix, 100)
        else:
            self.tqdm = tqdm(iterable, self.prefix, leave=False, disable=False)

    def __iter__(self):
        return iter(self._fallback if self.tqdm is None else self.tqdm)

    def log(self, stats, tag=None, step=None) -> None:
        if self.tqdm is None:
            return self._fallback.log(stats, tag, step)
        self.tqdm.set_postfix(self._format_stats(stats), refresh=False)

    def print(self, stats, tag=None, step=None) -> None:
        body = self._str_pipes(self._format_stats(stats))
        with rename_logger(logger, tag) as sink:
            sink.info(f"{self.prefix} | {body}")


_BAR_KINDS = {
    "json": lambda it, ep, pre, ival: JsonProgressBar(it, ep, pre, ival),
    "none": lambda it, ep, pre, ival: NoopProgressBar(it, ep, pre),
    "simple": lambda it, ep, pre, ival: SimpleProgressBar(it, ep, pre, ival),
    "tqdm": lambda it, ep, pre, ival: TqdmProgressBar(it, ep, pre),
}


def progress_bar(
    iterator,
    log_format: Optional[str] = None,
    log_interval: int = 100,
    epoch: Optional[int] = None,
    prefix: Optional[str] = None,
    tensorboard_logdir: Optional[str] = None,
    default_log_format: str = "tqdm",
    wandb_project: Optional[str] = None,
    args=None,
):
    """Build the configured bar, optionally wrapped by the TB/wandb sink."""
    kind = log_format if log_format is not None else default_log_format
    if kind == "tqdm" and not sys.stderr.isatty():
        kind = "simple"  # tqdm redraws are noise in captured logs
    try:
        bar = _BAR_KINDS[kind](iterator, epoch, prefix, log_interval)
    except KeyError:
        raise ValueError(f"Unknown log format: {kind}")
    if tensorboard_logdir:
        bar = TensorboardProgressBarWrapper(
            bar, tensorboard_logdir, wandb_project, args
        )
    return bar


class TensorboardProgressBarWrapper(BaseProgressBar):
    """Mirrors numeric stats into tensorboard (and wandb) around an inner bar."""

    def __init__(self, wrapped_bar, tensorboard_logdir, wandb_project=None,
                 args=None):
        self.wrapped_bar = wrapped_bar
        self.tensorboard_logdir = tensorboard_logdir
        self._writers = {}
        self.wandb = None
        if wandb_project:
            try:
                import wandb

                wandb.init(project=wandb_project,
                           config=vars(args) if args else None)
                self.wandb = wandb
            except ImportError:
                logger.warning("wandb not found; skipping wandb logging")
        self.SummaryWriter = self._locate_summary_writer()

    @staticmethod
    def _locate_summary_writer():
        for modname in ("torch.utils.tensorboard", "tensorboardX"):
            try:
                mod = __import__(modname, fromlist=["SummaryWriter"])
                return mod.SummaryWriter
            except ImportError:
                continue
        logger.warning(
            "tensorboard not found; please install with: pip install tensorboard"
        )
        return None

    def _writer(self, key):
        if self.SummaryWriter is None:
            return None
        if key not in self._writers:
            w = self.SummaryWriter(os.path.join(self.tensorboard_logdir, key))
            w.add_text("sys.argv", " ".join(sys.argv))
            self._writers[key] = w
        return self._writers[key]

    def __len__(self):
        return len(self.wrapped_bar)

    def __iter__(self):
        return iter(self.wrapped_bar)

    def log(self, stats, tag=None, step=None) -> None:
        self._emit(stats, tag, step)
        self.wrapped_bar.log(stats, tag=tag, step=step)

    def print(self, stats, tag=None, step=None) -> None:
        self._emit(stats, tag, step)
        self.wrapped_bar.print(stats, tag=tag, step=step)

    def _emit(self, stats, tag=None, step=None):
        writer = self._writer(tag or "")
        if writer is None and self.wandb is None:
            return
        if step is None:
            step = stats.get("num_updates", 0) if hasattr(stats, "get") else 0
        for key in stats.keys() - {"num_updates"}:
            value = stats[key]
            if isinstance(value, AverageMeter):
                value = value.val
            elif not isinstance(value, Number):
                continue
            if writer is not None:
                writer.add_scalar(key, value, step)
            if self.wandb is not None:
                ns = f"{tag}/" if tag else ""
                self.wandb.log({ns + key: value}, step=step)
        if writer is not None:
            writer.flush()
