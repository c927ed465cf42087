from .scheduler import ContinuousBatcher, Request  # noqa: F401
