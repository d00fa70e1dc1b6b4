from .model_reward import ModelReward  # noqa: F401
from .rule_math import MathRuleReward, constant_reward  # noqa: F401
