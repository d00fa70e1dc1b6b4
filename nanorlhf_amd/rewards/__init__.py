from .model_reward import ModelReward  # noqa: F401
from .string_reward import StringReward  # noqa: F401
from .rule_math import MathRuleReward, constant_reward  # noqa: F401
