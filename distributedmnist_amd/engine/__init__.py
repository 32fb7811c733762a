from .supervisor import Supervisor  # noqa: F401
from .train import Trainer, train_main, lr_at  # noqa: F401
from . import evaluate  # noqa: F401
