from .lenet import LeNet5  # noqa: F401
from .mlp import MLP  # noqa: F401


def build_model(name: str, **kw):
    if name in ("lenet", "cnn", "lenet5"):
        return LeNet5(**kw)
    if name == "mlp":
        return MLP(**kw)
    raise ValueError(f"unknown model {name!r} (expected 'lenet' or 'mlp')")
