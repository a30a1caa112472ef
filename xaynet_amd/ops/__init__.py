from .engine import GpuMaskedAggregator, gpu_available  # noqa: F401


def make_coordinator_driver(coordinator, vect_cfg, unit_cfg, length, **kw):
    """Lazy import (torch) of the staged-plane GPU driver."""
    from .driver import GpuCoordinatorDriver

    return GpuCoordinatorDriver(coordinator, vect_cfg, unit_cfg, length, **kw)
