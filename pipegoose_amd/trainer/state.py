"""Trainer lifecycle state (reference parity: pipegoose/trainer/state.py:4-19)."""
from dataclasses import dataclass, field
from enum import Enum


class TrainerStage(Enum):
    IDLE = "idle"
    TRAINING = "training"
    EVALUATING = "evaluating"
    FINISHED = "finished"


@dataclass
class TrainerState:
    stage: TrainerStage = TrainerStage.IDLE
    epoch: int = 0
    global_step: int = 0
    samples_seen: int = 0
    tokens_seen: int = 0
    last_loss: float = float("nan")
    metrics: dict = field(default_factory=dict)
