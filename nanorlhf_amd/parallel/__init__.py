from . import dist  # noqa: F401
from .ddp import GradReducer  # noqa: F401
