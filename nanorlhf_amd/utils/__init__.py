from .seed import rank_seed, set_seed  # noqa: F401
from .checkpoint import CheckpointManager  # noqa: F401
from .logging import MetricsLogger  # noqa: F401
from .offload import OffloadEngine  # noqa: F401
from .timers import PhaseTimers  # noqa: F401
