from .config import ModelConfig, get_config, PRESETS  # noqa: F401
from .qwen2 import AttnContext, CausalLM, Transformer, make_positions, pack_sequences  # noqa: F401
from .value_head import ScalarHeadModel  # noqa: F401
from .lora import LoraConfig, LoRALinear, apply_lora, merge_for_rollout, unmerge  # noqa: F401
