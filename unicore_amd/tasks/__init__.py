"""Task registry (parity: reference unicore/tasks/__init__.py:16-61).

Tasks self-register at import time via :func:`register_task`; every module
in this directory is imported below so built-in tasks appear without any
explicit list.
"""

import importlib
import os

from .unicore_task import UnicoreTask

TASK_REGISTRY = {}
TASK_CLASS_NAMES = set()


def setup_task(args, **kwargs):
    return TASK_REGISTRY[args.task].setup_task(args, **kwargs)


def register_task(name):
    """Decorator registering a UnicoreTask subclass under *name*::

        @register_task('classification')
        class ClassificationTask(UnicoreTask):
            ...
    """

    def wrap(cls):
        if name in TASK_REGISTRY:
            raise ValueError(f"Cannot register duplicate task ({name})")
        if not issubclass(cls, UnicoreTask):
            raise ValueError(
                f"Task ({name}: {cls.__name__}) must extend UnicoreTask"
            )
        if cls.__name__ in TASK_CLASS_NAMES:
            raise ValueError(
                f"Cannot register task with duplicate class name "
                f"({cls.__name__})"
            )
        TASK_REGISTRY[name] = cls
        TASK_CLASS_NAMES.add(cls.__name__)
        return cls

    return wrap


def get_task(name):
    return TASK_REGISTRY[name]


def _import_all_task_modules():
    here = os.path.dirname(__file__)
    for entry in sorted(os.listdir(here)):
        if entry.startswith(("_", ".")):
            continue
        is_pkg = os.path.isdir(os.path.join(here, entry))
        if not (entry.endswith(".py") or is_pkg):
            continue
        modname = entry[:-3] if entry.endswith(".py") else entry
        if modname != "unicore_task":
            importlib.import_module(f"unicore_amd.tasks.{modname}")


_import_all_task_modules()
