"""`AsyncParticipant` — API-compatible with the reference's
bindings/python/xaynet_sdk/async_participant.py:15-119: a queue-style handle
(get_global_model / set_local_model) driven by a background tick thread, with
a threading.Event notifier for new global models.
"""
from __future__ import annotations

import logging
import threading
from typing import List, Optional

from . import xaynet_sdk
from .participant import _Backoff

xaynet_sdk.init_logging()
LOG = logging.getLogger("participant")


class AsyncParticipant(threading.Thread):
    def __init__(self, coordinator_url, notifier, state, scalar):
        self._xaynet_participant = xaynet_sdk.Participant(coordinator_url, scalar, state)
        self._exit_event = threading.Event()
        self._poll_period = _Backoff(min_ms=100, max_ms=10000, factor=1.2)
        self._notifier = notifier
        self._tick_lock = threading.Lock()
        super().__init__(daemon=True)

    def run(self):
        try:
            while not self._exit_event.is_set():
                self._tick()
        except Exception as err:  # noqa: BLE001 — reference parity
            LOG.error("unrecoverable error: %s shut down participant", err)
            self._exit_event.set()

    def _tick(self):
        with self._tick_lock:
            self._xaynet_participant.tick()
            new_global_model = self._xaynet_participant.new_global_model()
            made_progress = self._xaynet_participant.made_progress()

        if new_global_model and not self._notifier.is_set():
            LOG.debug("notify that a new global model is available")
            self._notifier.set()

        if made_progress:
            self._poll_period.reset()
        self._exit_event.wait(timeout=self._poll_period.duration())

    def get_global_model(self) -> Optional[list]:
        """Current global model as a list (None when no model exists yet)."""
        LOG.debug("get global model")
        self._notifier.clear()
        with self._tick_lock:
            return self._xaynet_participant.global_model()

    def set_local_model(self, local_model: list):
        """Cache a local model for the next update task."""
        LOG.debug("set local model in model store")
        with self._tick_lock:
            self._xaynet_participant.set_model(local_model)

    def stop(self) -> List[int]:
        """Stop the thread and return the serialized participant state."""
        LOG.debug("stop participant")
        self._exit_event.set()
        self._notifier.clear()
        with self._tick_lock:
            return self._xaynet_participant.save()
