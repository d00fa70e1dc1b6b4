from .synthetic import PromptDataset, hh_shaped_prompts, math_shaped_prompts  # noqa: F401
from .buckets import create_batches  # noqa: F401
