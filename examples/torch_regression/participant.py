"""Federated linear regression with PyTorch — the ML-framework-integration
example (the reference ships a Keras house-prices regression; this is the
torch equivalent on synthetic data).

Each participant owns a private data shard; `train_round` does a few SGD
epochs starting from the global model and returns the updated weights.

Start a coordinator configured for the right model length first, e.g.:
    XAYNET__MODEL__LENGTH=11 python -m xaynet_amd.server -c configs/config.toml
(10 features + bias = 11 weights), then run several of these.
"""
import argparse
import logging
from typing import Optional

import numpy as np
import torch

import xaynet_sdk

logging.basicConfig(level=logging.INFO)

N_FEATURES = 10
TRUE_W = np.linspace(-0.5, 0.5, N_FEATURES)


class RegressionParticipant(xaynet_sdk.ParticipantABC):
    def __init__(self, seed: int, samples: int = 256):
        rng = np.random.default_rng(seed)
        x = rng.normal(size=(samples, N_FEATURES)).astype(np.float32)
        noise = rng.normal(scale=0.05, size=samples).astype(np.float32)
        y = (x @ TRUE_W.astype(np.float32) + 0.1 + noise).astype(np.float32)
        self.x = torch.from_numpy(x)
        self.y = torch.from_numpy(y)
        self.model = torch.nn.Linear(N_FEATURES, 1)

    # ---- xaynet_sdk.ParticipantABC ----

    def deserialize_training_input(self, global_model: list):
        return torch.tensor(global_model, dtype=torch.float32)

    def train_round(self, training_input: Optional[torch.Tensor]):
        if training_input is not None:
            with torch.no_grad():
                self.model.weight.copy_(training_input[:N_FEATURES].view(1, -1))
                self.model.bias.copy_(training_input[N_FEATURES:])
        opt = torch.optim.SGD(self.model.parameters(), lr=0.05)
        for _ in range(5):
            opt.zero_grad()
            loss = torch.nn.functional.mse_loss(self.model(self.x).squeeze(-1), self.y)
            loss.backward()
            opt.step()
        logging.info("local loss after training: %.5f", float(loss.detach()))
        return self.model

    def serialize_training_result(self, training_result) -> list:
        w = training_result.weight.detach().view(-1)
        b = training_result.bias.detach().view(-1)
        return torch.cat([w, b]).clamp(-1, 1).tolist()  # B0 bound

    def on_new_global_model(self, global_model) -> None:
        if global_model is not None:
            err = float(
                np.abs(np.asarray(global_model[:N_FEATURES]) - TRUE_W).mean()
            )
            logging.info("global model mean |w - w*| = %.5f", err)


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--url", default="http://127.0.0.1:8081")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    handle = xaynet_sdk.spawn_participant(
        args.url, RegressionParticipant, args=(args.seed,)
    )
    try:
        handle.join()
    except KeyboardInterrupt:
        handle.stop()
