from . import meters, metrics, progress_bar  # noqa: F401
