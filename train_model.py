"""Drop-in root module matching the reference layout
(/root/reference/train_model.py): `from train_model import train,
validate, test` works as it does against the reference. The real
implementations live in factorvae_amd.engine.trainer (the fused-engine
epoch loops are in factorvae_amd.engine.fused).
"""
from factorvae_amd.engine.trainer import test, train, train_main, validate

__all__ = ["train", "validate", "test", "train_main"]
