"""Watch for new global models with an AsyncParticipant (no training):
the notifier event fires when a round publishes a model (reference
bindings/python/examples/download_global_model_async.py).
"""
import xaynet_sdk


def main():
    participant, new_model = xaynet_sdk.spawn_async_participant("http://127.0.0.1:8081")
    try:
        while True:
            if new_model.wait(timeout=10.0):
                model = participant.get_global_model()
                if model is not None:
                    print(f"new global model ({len(model)} weights): {model[:4]}...")
    except KeyboardInterrupt:
        participant.stop()


if __name__ == "__main__":
    main()
