"""Training timestamp — Composer Timestamp-equivalent counters.

Tracks epoch/batch/sample/token counts; the federated round loop copies and
restores these across rounds (reference llm_client_functions.py:163-175)."""

from __future__ import annotations

from dataclasses import asdict, dataclass


@dataclass
class Timestamp:
    epoch: int = 0
    batch: int = 0
    sample: int = 0
    token: int = 0

    def tick_batch(self, samples: int, tokens: int) -> None:
        self.batch += 1
        self.sample += samples
        self.token += tokens

    def state_dict(self) -> dict:
        return asdict(self)

    def load_state_dict(self, state: dict) -> None:
        for k, v in state.items():
            setattr(self, k, int(v))

    def copy(self) -> "Timestamp":
        return Timestamp(**asdict(self))
