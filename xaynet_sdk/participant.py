"""`ParticipantABC` + `InternalParticipant` — API-compatible with the
reference's bindings/python/xaynet_sdk/participant.py:19-226.

The user subclasses `ParticipantABC` (train_round / serialize_training_result
/ deserialize_training_input + optional callbacks); `InternalParticipant` is a
daemon thread driving the native participant's tick loop with exponential
backoff polling (100 ms -> 10 s, factor 1.2).
"""
from __future__ import annotations

import logging
import threading
from abc import ABC, abstractmethod
from typing import List, Optional, TypeVar

from . import xaynet_sdk

xaynet_sdk.init_logging()
LOG = logging.getLogger("participant")

TrainingResult = TypeVar("TrainingResult")
TrainingInput = TypeVar("TrainingInput")


class _Backoff:
    """Minimal exponential backoff (the reference uses justbackoff)."""

    def __init__(self, min_ms=100.0, max_ms=10000.0, factor=1.2):
        self.min_s = min_ms / 1000.0
        self.max_s = max_ms / 1000.0
        self.factor = factor
        self._attempts = 0

    def duration(self) -> float:
        d = self.min_s * (self.factor**self._attempts)
        self._attempts += 1
        return min(d, self.max_s)

    def reset(self):
        self._attempts = 0


class ParticipantABC(ABC):
    @abstractmethod
    def train_round(self, training_input: Optional[TrainingInput]) -> TrainingResult:
        """Train a local model. `training_input` is the deserialized global
        model, or None when no global model exists yet (first round). Returns
        the updated local model."""
        raise NotImplementedError()

    @abstractmethod
    def serialize_training_result(self, training_result: TrainingResult) -> list:
        """Serialize `training_result` into a list whose element type matches
        the coordinator's configured data type."""
        raise NotImplementedError()

    @abstractmethod
    def deserialize_training_input(self, global_model: list) -> TrainingInput:
        """Deserialize the global model list into the `TrainingInput` that
        `train_round` consumes."""
        raise NotImplementedError()

    def participate_in_update_task(self) -> bool:
        """Whether `train_round` should run when selected as updater."""
        return True

    def on_new_global_model(self, global_model: Optional[TrainingInput]) -> None:
        """Called when a new global model is available."""

    def on_stop(self) -> None:
        """Called before the participant thread stops."""


class InternalParticipant(threading.Thread):
    def __init__(self, coordinator_url, participant, p_args, p_kwargs, state, scalar,
                 gpu=False):
        self._xaynet_participant = xaynet_sdk.Participant(coordinator_url, scalar, state)
        if gpu:
            from xaynet_amd.ops.accel import ParticipantAccel

            # shared process-wide accelerator (engines cached per round config)
            global _ACCEL
            if "_ACCEL" not in globals() or _ACCEL is None:
                _ACCEL = ParticipantAccel()
            _ACCEL.attach(self._xaynet_participant._inner)

        # the user participant is constructed on the thread (run()) so the ML
        # model lives on the participant thread (reference participant.py:129-149)
        self._participant = participant
        self._p_args = tuple(p_args)
        self._p_kwargs = dict(p_kwargs)

        self._exit_event = threading.Event()
        self._poll_period = _Backoff(min_ms=100, max_ms=10000, factor=1.2)

        self._global_model = None
        self._error_on_fetch_global_model = False

        self._tick_lock = threading.Lock()

        super().__init__(daemon=True)

    def run(self):
        self._participant = self._participant(*self._p_args, **self._p_kwargs)
        try:
            while not self._exit_event.is_set():
                self._tick()
        except Exception as err:  # noqa: BLE001 — reference parity
            LOG.error("unrecoverable error: %s shut down participant", err)
            self._exit_event.set()

    def _fetch_global_model(self):
        LOG.debug("fetch global model")
        try:
            global_model = self._xaynet_participant.global_model()
        except (
            xaynet_sdk.GlobalModelUnavailable,
            xaynet_sdk.GlobalModelDataTypeMisMatch,
        ) as err:
            LOG.warning("failed to get global model: %s", err)
            self._error_on_fetch_global_model = True
        else:
            if global_model is not None:
                self._global_model = self._participant.deserialize_training_input(
                    global_model
                )
            else:
                self._global_model = None
            self._error_on_fetch_global_model = False

    def _train(self):
        LOG.debug("train model")
        data = self._participant.train_round(self._global_model)
        local_model = self._participant.serialize_training_result(data)
        try:
            self._xaynet_participant.set_model(local_model)
        except (
            xaynet_sdk.LocalModelLengthMisMatch,
            xaynet_sdk.LocalModelDataTypeMisMatch,
        ) as err:
            LOG.warning("failed to set local model: %s", err)

    def _tick(self):
        with self._tick_lock:
            self._xaynet_participant.tick()

            if (
                self._xaynet_participant.new_global_model()
                or self._error_on_fetch_global_model
            ):
                self._fetch_global_model()
                if not self._error_on_fetch_global_model:
                    self._participant.on_new_global_model(self._global_model)

            if (
                self._xaynet_participant.should_set_model()
                and self._participant.participate_in_update_task()
                and not self._error_on_fetch_global_model
            ):
                self._train()

            made_progress = self._xaynet_participant.made_progress()

        if made_progress:
            self._poll_period.reset()
        self._exit_event.wait(timeout=self._poll_period.duration())

    def stop(self) -> List[int]:
        """Stop the thread and return the serialized participant state
        (consumes the participant; reference participant.py:219-226)."""
        LOG.debug("stop participant")
        self._exit_event.set()
        with self._tick_lock:
            state = self._xaynet_participant.save()
        if hasattr(self._participant, "on_stop"):
            try:
                self._participant.on_stop()
            except Exception:  # noqa: BLE001
                pass
        return state
