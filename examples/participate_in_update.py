"""Update-task-only participation: skip training when resources are tight by
overriding `participate_in_update_task` (reference
bindings/python/examples/participate_in_update.py).

Start a coordinator first: python -m xaynet_amd.server -c configs/config.toml
"""
import logging
import random
from typing import Optional

import xaynet_sdk

logging.basicConfig(level=logging.INFO)


class SometimesParticipant(xaynet_sdk.ParticipantABC):
    """Participates in ~50% of the update tasks it is selected for (e.g. only
    when the device is charging)."""

    def participate_in_update_task(self) -> bool:
        go = random.random() < 0.5
        if not go:
            print("selected as updater, but skipping this round")
        return go

    def train_round(self, training_input: Optional[list]) -> list:
        return [0.4] * 4

    def serialize_training_result(self, training_result) -> list:
        return list(training_result)

    def deserialize_training_input(self, global_model: list) -> list:
        return global_model


if __name__ == "__main__":
    handle = xaynet_sdk.spawn_participant("http://127.0.0.1:8081", SometimesParticipant)
    try:
        handle.join()
    except KeyboardInterrupt:
        handle.stop()
