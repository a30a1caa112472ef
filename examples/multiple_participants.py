"""Run several participants from one process (each gets its own thread and
its own PET identity) — handy for local protocol testing.

For serious load generation use scripts/test_drive.py instead.
Start a coordinator first: python -m xaynet_amd.server -c configs/config.toml
"""
import logging
import time
from typing import Optional

import xaynet_sdk

logging.basicConfig(level=logging.INFO)

N = 5


class Member(xaynet_sdk.ParticipantABC):
    def __init__(self, idx: int):
        self.idx = idx

    def train_round(self, training_input: Optional[list]) -> list:
        return [float(self.idx)] * 4

    def serialize_training_result(self, training_result) -> list:
        return list(training_result)

    def deserialize_training_input(self, global_model: list) -> list:
        return global_model

    def on_new_global_model(self, global_model: Optional[list]) -> None:
        if global_model is not None:
            print(f"[{self.idx}] sees global model {global_model}")


if __name__ == "__main__":
    handles = [
        xaynet_sdk.spawn_participant("http://127.0.0.1:8081", Member, args=(i,))
        for i in range(N)
    ]
    try:
        while True:
            time.sleep(1)
    except KeyboardInterrupt:
        for h in handles:
            h.stop()
