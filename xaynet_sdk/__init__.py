"""Drop-in replacement for the reference's `xaynet_sdk` Python package
(bindings/python/xaynet_sdk/__init__.py): spawn_participant /
spawn_async_participant on top of the xaynet_amd native participant.
"""
import threading
from typing import List, Optional, Tuple

from .async_participant import AsyncParticipant
from .participant import InternalParticipant, ParticipantABC

__all__ = [
    "ParticipantABC",
    "InternalParticipant",
    "AsyncParticipant",
    "spawn_participant",
    "spawn_async_participant",
]


def spawn_participant(
    coordinator_url: str,
    participant: type,
    args: Tuple = (),
    kwargs: dict = {},
    state: Optional[List[int]] = None,
    scalar: float = 1.0,
    gpu: bool = False,
):
    """Spawn an `InternalParticipant` thread and return its handle. If `state`
    is given, the participant is restored from it. `scalar` weights this
    participant's update in the aggregate (e.g. 1/number_of_samples).
    `gpu=True` runs the update-masking and sum2 mask-aggregation hot loops
    on a visible MI355X (falls back to CPU per call on any error)."""
    internal_participant = InternalParticipant(
        coordinator_url, participant, args, kwargs, state, scalar, gpu=gpu
    )
    internal_participant.start()
    return internal_participant


def spawn_async_participant(
    coordinator_url: str, state: Optional[List[int]] = None, scalar: float = 1.0
):
    """Spawn an `AsyncParticipant` thread; returns (participant, notifier)
    where `notifier` is a threading.Event set when a new global model is
    available."""
    notifier = threading.Event()
    async_participant = AsyncParticipant(coordinator_url, notifier, state, scalar)
    async_participant.start()
    return (async_participant, notifier)
