from .trainer import Trainer, train_from_config  # noqa: F401
