"""Generic plugin registries.

Behavioral parity with the reference's ``setup_registry`` factory
(reference: unicore/registry.py:13-81): each registry exposes a
``register_x`` decorator and a ``build_x(args, *extra)`` factory, registered
classes may contribute CLI args via a classmethod ``add_args(parser)``, and
``--x`` choices are derived from the registry keys. ``REGISTRIES`` indexes
every registry so options.py can install one choice flag per registry.
"""

import argparse

REGISTRIES = {}


def setup_registry(registry_name: str, base_class=None, default=None,
                   required=False):
    assert registry_name.startswith("--")
    key = registry_name[2:].replace("-", "_")

    if key in REGISTRIES:
        raise ValueError(f"Cannot setup duplicate registry: {key}")
    table = {}
    seen_class_names = set()
    REGISTRIES[key] = {"registry": table, "default": default}

    def build_x(args, *extra_args, **extra_kwargs):
        choice = getattr(args, key, None)
        if choice is None:
            if required:
                raise ValueError(f"--{key} is required")
            return None
        chosen = table[choice]
        # classes may define build_<key> as an alternate constructor
        factory = getattr(chosen, "build_" + key, chosen)
        set_defaults(args, chosen)
        return factory(args, *extra_args, **extra_kwargs)

    def register_x(name):
        def wrap(cls):
            if name in table:
                raise ValueError(f"Cannot register duplicate {key} ({name})")
            if cls.__name__ in seen_class_names:
                raise ValueError(
                    f"Cannot register {key} with duplicate class name "
                    f"({cls.__name__})"
                )
            if base_class is not None and not issubclass(cls, base_class):
                raise ValueError(
                    f"{cls.__name__} must extend {base_class.__name__}"
                )
            table[name] = cls
            seen_class_names.add(cls.__name__)
            return cls

        return wrap

    return build_x, register_x, table


def set_defaults(args, cls):
    """Backfill args with the defaults cls.add_args declares, for any
    attribute the user never set (reference unicore/registry.py:66-81)."""
    if not hasattr(cls, "add_args"):
        return
    probe = argparse.ArgumentParser(
        argument_default=argparse.SUPPRESS, allow_abbrev=False
    )
    cls.add_args(probe)
    for action in probe._actions:
        dest = action.dest
        if dest is argparse.SUPPRESS or action.default is argparse.SUPPRESS:
            continue
        if not hasattr(args, dest):
            setattr(args, dest, action.default)
