from .cbow import CbowTrainer, TrainResult  # noqa: F401
