from .flatten import FlatParams  # noqa: F401
from .sync import SyncEngine, MODES  # noqa: F401
from .timers import StepTimer  # noqa: F401
