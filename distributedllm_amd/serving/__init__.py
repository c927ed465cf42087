from .scheduler import ContinuousBatcher, Request  # noqa: F401
from .speculative import SpecStats, pld_generate  # noqa: F401


def build_http_app(batcher, tokenizer, eos_id=None):
    """Lazy import so fastapi is only required for HTTP serving."""
    from .http import build_app
    return build_app(batcher, tokenizer, eos_id=eos_id)
