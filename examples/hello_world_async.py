"""Queue-style participation with `AsyncParticipant`: the app pushes local
models and polls for new global models — no callback subclassing.

Start a coordinator first: python -m xaynet_amd.server -c configs/config.toml
"""
import logging

import xaynet_sdk

logging.basicConfig(level=logging.INFO)


def main():
    participant, new_model_event = xaynet_sdk.spawn_async_participant(
        "http://127.0.0.1:8081"
    )
    local_model = [0.25, 0.5, 0.75, 1.0]
    try:
        while True:
            participant.set_local_model(local_model)
            if new_model_event.wait(timeout=5.0):
                global_model = participant.get_global_model()
                print(f"global model now: {global_model}")
    except KeyboardInterrupt:
        participant.stop()


if __name__ == "__main__":
    main()
