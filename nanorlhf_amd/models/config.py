"""Model configurations and named presets.

Presets cover the model families the reference runs (SURVEY.md §6 /
BASELINE.json configs): Qwen2.5-{0.5,1.5,7}B policy shapes, a
deberta-v3-large-SHAPED bidirectional reward encoder (same parameter count /
layer geometry; MI355X-native architecture, not a DeBERTa port), and the
2-layer 128-dim tiny model for the CPU plumbing config."""
from __future__ import annotations

from dataclasses import dataclass, field, asdict


@dataclass
class ModelConfig:
    vocab_size: int = 151936
    hidden_size: int = 1536
    num_layers: int = 28
    num_heads: int = 12
    num_kv_heads: int = 2
    head_dim: int = 128
    intermediate_size: int = 8960
    rms_eps: float = 1e-6
    rope_theta: float = 1e6
    max_position: int = 32768
    tie_word_embeddings: bool = True
    qkv_bias: bool = True
    dtype: str = "bfloat16"
    # encoder-only (reward model) options
    bidirectional: bool = False
    num_labels: int = 0  # >0 adds a classification/score head

    def to_dict(self):
        return asdict(self)

    @staticmethod
    def from_dict(d):
        return ModelConfig(**d)


PRESETS: dict[str, dict] = {
    # BASELINE config #1: 2-layer 128-dim tiny model, CPU-only plumbing.
    "tiny": dict(vocab_size=1024, hidden_size=128, num_layers=2, num_heads=4,
                 num_kv_heads=2, head_dim=32, intermediate_size=512,
                 max_position=2048, rope_theta=1e4, dtype="float32"),
    # Qwen2.5 shapes (random-init; no network for checkpoints).
    "qwen2.5-0.5b": dict(vocab_size=151936, hidden_size=896, num_layers=24,
                         num_heads=14, num_kv_heads=2, head_dim=64,
                         intermediate_size=4864, tie_word_embeddings=True),
    "qwen2.5-1.5b": dict(vocab_size=151936, hidden_size=1536, num_layers=28,
                         num_heads=12, num_kv_heads=2, head_dim=128,
                         intermediate_size=8960, tie_word_embeddings=True),
    "qwen2.5-7b": dict(vocab_size=152064, hidden_size=3584, num_layers=28,
                       num_heads=28, num_kv_heads=4, head_dim=128,
                       intermediate_size=18944, tie_word_embeddings=False),
    # deberta-v3-large-shaped reward encoder (24L x 1024h, ~400M backbone).
    "rm-large": dict(vocab_size=128100, hidden_size=1024, num_layers=24,
                     num_heads=16, num_kv_heads=16, head_dim=64,
                     intermediate_size=4096, bidirectional=True, num_labels=1,
                     rope_theta=1e4, max_position=4096, tie_word_embeddings=False,
                     qkv_bias=True),
    # tiny reward encoder for CPU tests
    "rm-tiny": dict(vocab_size=1024, hidden_size=128, num_layers=2, num_heads=4,
                    num_kv_heads=4, head_dim=32, intermediate_size=512,
                    bidirectional=True, num_labels=1, rope_theta=1e4,
                    max_position=2048, dtype="float32", tie_word_embeddings=False),
}


def get_config(name: str, **overrides) -> ModelConfig:
    if name not in PRESETS:
        raise KeyError(f"unknown model preset {name!r}; have {sorted(PRESETS)}")
    d = dict(PRESETS[name])
    d.update(overrides)
    return ModelConfig(**d)
