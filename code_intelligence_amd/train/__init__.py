from .trainer import LMTrainer, TrainConfig
from .callbacks import (Callback, CallbackList, EarlyStopping, SaveModel,
                        ReduceLROnPlateau, CSVLogger, JSONRunLogger,
                        TerminateOnNaN)
from .schedules import OneCycle, FlatSchedule

__all__ = [
    "LMTrainer", "TrainConfig", "Callback", "CallbackList", "EarlyStopping",
    "SaveModel", "ReduceLROnPlateau", "CSVLogger", "JSONRunLogger",
    "TerminateOnNaN", "OneCycle", "FlatSchedule",
]
