from .fm import FMModel, FMTrainer

__all__ = ["FMModel", "FMTrainer"]
