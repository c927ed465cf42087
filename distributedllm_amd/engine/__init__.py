from .slice_engine import (  # noqa: F401
    HIPSliceEngine, TorchSliceEngine, engine_for_slice)
