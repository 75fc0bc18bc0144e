"""Generic plugin registries.

Behavioral parity with the reference's ``setup_registry`` factory
(reference: unicore/registry.py:13-81): each registry exposes a
``register_x`` decorator and a ``build_x(args, *extra)`` factory, registered
classes may contribute CLI args via a classmethod ``add_args(parser)``, and
``--x`` choices are derived from the registry keys.
"""

import argparse

REGISTRIES = {}


def setup_registry(registry_name: str, base_class=None, default=None, required=False):
    assert registry_name.startswith("--")
    clean_name = registry_name[2:].replace("-", "_")

    REGISTRY = {}
    REGISTRY_CLASS_NAMES = set()

    # maintain a registry of all registries
    if clean_name in REGISTRIES:
        raise ValueError(f"Cannot setup duplicate registry: {clean_name}")
    REGISTRIES[clean_name] = {"registry": REGISTRY, "default": default}

    def build_x(args, *extra_args, **extra_kwargs):
        choice = getattr(args, clean_name, None)
        if choice is None:
            if required:
                raise ValueError(f"--{clean_name} is required")
            return None
        cls = REGISTRY[choice]
        if hasattr(cls, "build_" + clean_name):
            builder = getattr(cls, "build_" + clean_name)
        else:
            builder = cls
        set_defaults(args, cls)
        return builder(args, *extra_args, **extra_kwargs)

    def register_x(name):
        def register_x_cls(cls):
            if name in REGISTRY:
                raise ValueError(f"Cannot register duplicate {clean_name} ({name})")
            if cls.__name__ in REGISTRY_CLASS_NAMES:
                raise ValueError(
                    f"Cannot register {clean_name} with duplicate class name ({cls.__name__})"
                )
            if base_class is not None and not issubclass(cls, base_class):
                raise ValueError(
                    f"{cls.__name__} must extend {base_class.__name__}"
                )
            REGISTRY[name] = cls
            REGISTRY_CLASS_NAMES.add(cls.__name__)
            return cls

        return register_x_cls

    return build_x, register_x, REGISTRY


def set_defaults(args, cls):
    """Apply defaults from cls.add_args to any attribute not already set.

    Mirrors the behavior at reference unicore/registry.py:66-81: a throwaway
    parser collects the class's declared defaults, and any arg the user did
    not supply inherits them.
    """
    if not hasattr(cls, "add_args"):
        return
    parser = argparse.ArgumentParser(argument_default=argparse.SUPPRESS, allow_abbrev=False)
    cls.add_args(parser)
    defaults = argparse.Namespace()
    for action in parser._actions:
        if action.dest is not argparse.SUPPRESS:
            if not hasattr(defaults, action.dest):
                if action.default is not argparse.SUPPRESS:
                    setattr(defaults, action.dest, action.default)
    for key, default_value in vars(defaults).items():
        if not hasattr(args, key):
            setattr(args, key, default_value)
