from .metrics import StageTimer, ThroughputMeter  # noqa: F401
