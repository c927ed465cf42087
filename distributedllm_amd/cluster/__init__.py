from . import client, node, protocol, uploads  # noqa: F401
