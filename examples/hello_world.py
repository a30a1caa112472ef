"""Minimal participant: joins rounds and contributes a fixed weight vector.

Demonstrates the `xaynet_sdk.ParticipantABC` surface (API-compatible with the
reference Python SDK). Start a coordinator first, e.g.:

    python -m xaynet_amd.server -c configs/config.toml

then run this script once per simulated device.
"""
import logging
from typing import Optional

import xaynet_sdk

logging.basicConfig(level=logging.INFO)


class FixedModelParticipant(xaynet_sdk.ParticipantABC):
    """Contributes the same 4-weight vector every round it is selected."""

    WEIGHTS = [0.1, 0.2, 0.345, 0.3]

    def train_round(self, training_input: Optional[list]) -> list:
        # a real app would fine-tune on local data here, seeded from
        # `training_input` (the current global model, None in round 1)
        return self.WEIGHTS

    def serialize_training_result(self, training_result) -> list:
        return list(training_result)

    def deserialize_training_input(self, global_model: list) -> list:
        return global_model

    def on_new_global_model(self, global_model: Optional[list]) -> None:
        if global_model is not None:
            print(f"new global model: {global_model}")


if __name__ == "__main__":
    handle = xaynet_sdk.spawn_participant("http://127.0.0.1:8081", FixedModelParticipant)
    try:
        handle.join()
    except KeyboardInterrupt:
        handle.stop()
