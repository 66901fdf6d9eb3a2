from .trainer import train, validate, test, train_main

__all__ = ["train", "validate", "test", "train_main"]
