from . import ggml, q4, slicer  # noqa: F401
