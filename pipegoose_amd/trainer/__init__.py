from pipegoose_amd.trainer.callback import Callback
from pipegoose_amd.trainer.logger import DistributedLogger, ThroughputMeter
from pipegoose_amd.trainer.state import TrainerStage, TrainerState
from pipegoose_amd.trainer.trainer import Trainer

__all__ = ["Trainer", "Callback", "DistributedLogger", "ThroughputMeter",
           "TrainerStage", "TrainerState"]
