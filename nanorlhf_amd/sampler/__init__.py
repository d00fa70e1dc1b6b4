from .cache import PagedKVCache, SeqState  # noqa: F401
from .engine import SamplerEngine, SamplingParams  # noqa: F401
