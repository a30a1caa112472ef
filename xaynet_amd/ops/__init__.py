from .engine import GpuMaskedAggregator, gpu_available  # noqa: F401
