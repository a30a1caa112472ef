"""Save a participant's state and restore it later (device restart survival).

The state blob contains the signing identity and the mid-round protocol
position; treat it as a secret. Works with both `spawn_participant` and
`spawn_async_participant` (the `state=` argument).

Start a coordinator first: python -m xaynet_amd.server -c configs/config.toml
"""
import logging
import pathlib
import time
from typing import Optional

import xaynet_sdk

logging.basicConfig(level=logging.INFO)
STATE_FILE = pathlib.Path("participant.state")


class Echo(xaynet_sdk.ParticipantABC):
    def train_round(self, training_input: Optional[list]) -> list:
        return [0.5] * 4

    def serialize_training_result(self, training_result) -> list:
        return list(training_result)

    def deserialize_training_input(self, global_model: list) -> list:
        return global_model


def main():
    state = list(STATE_FILE.read_bytes()) if STATE_FILE.exists() else None
    if state:
        print("restoring participant from", STATE_FILE)
    handle = xaynet_sdk.spawn_participant(
        "http://127.0.0.1:8081", Echo, state=state
    )
    time.sleep(15)  # ... participate for a while ...
    saved = handle.stop()
    STATE_FILE.write_bytes(bytes(saved))
    print("state saved to", STATE_FILE, "- run again to restore")


if __name__ == "__main__":
    main()
