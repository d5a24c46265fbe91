from .ddp import DPContext, HourglassStepTrainer

__all__ = ["DPContext", "HourglassStepTrainer"]
