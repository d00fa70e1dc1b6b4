from . import functional  # noqa: F401
from .trainer import AlgoSpec, RLHFTrainer, Rollout, TrainData  # noqa: F401
from .grpo import GRPO, GRPOConfig  # noqa: F401
from .ppo import PPO, PPOConfig  # noqa: F401
from .rloo import RLOO, RLOOConfig  # noqa: F401
from .remax import ReMax, RemaxConfig  # noqa: F401
from .raft import RAFT, RAFTConfig  # noqa: F401
from .reinforce import REINFORCE, ReinforceConfig  # noqa: F401
from .value_init import finetune_value_model  # noqa: F401
