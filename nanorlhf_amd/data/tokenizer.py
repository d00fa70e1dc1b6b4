"""Tokenizer integration + dataset prompt preparation.

The string-level seam the reference gets from transformers
(`/root/reference/GRPO/grpo.py:209-216` tokenizer load + [PAD],
:249-270 hh-rlhf first-human-turn extraction with the Qwen chat template;
`examples/r1-v0/grpo_r1.py:237-273` MetaMathQA question/answer records).
Only tokenization happens here — models never see strings; the trainer
stays on token ids and the reward adapter (rewards.StringReward) decodes
at the reward boundary, matching the reference's
`reward_func(pmt_and_responses: list[str])` contract.
"""
from __future__ import annotations

from typing import Iterable, Optional

# Fallback template for checkpoints without one (reference uses trl's
# SIMPLE_CHAT_TEMPLATE in the same situation, grpo.py:215-216).
SIMPLE_CHAT_TEMPLATE = (
    "{% for message in messages %}{{' ' + message['content']}}"
    "{% endfor %}{{eos_token}}"
)

# The literal prompt scaffold the reference wraps hh-rlhf questions in
# (grpo.py:252 — a plain string template, not tokenizer.apply_chat_template).
QWEN_CHAT_TEMPLATE = (
    "<|im_start|>system\nYou are Qwen, created by Alibaba Cloud. "
    "You are a helpful assistant.<|im_end|>\n"
    "<|im_start|>user\nQUESTION<|im_end|>\n<|im_start|>assistant\n"
)


def load_tokenizer(path: str, padding_side: str = "left"):
    """Load an HF tokenizer from a local directory and add the [PAD] token
    (reference grpo.py:209-216).  Returns the tokenizer; `len(tokenizer)`
    may exceed the model's vocab_size — the reference relies on the same
    slack rows at the end of the embedding."""
    from transformers import AutoTokenizer
    tok = AutoTokenizer.from_pretrained(path, padding_side=padding_side,
                                        trust_remote_code=False)
    tok.add_special_tokens({"pad_token": "[PAD]"})
    if getattr(tok, "chat_template", None) is None:
        tok.chat_template = SIMPLE_CHAT_TEMPLATE
    return tok


def extract_hh_question(chosen: str) -> str:
    """First human turn of an hh-rlhf `chosen` transcript (grpo.py:255-258)."""
    s = chosen.find("Human: ") + len("Human: ")
    e = chosen.find("Assistant: ", s)
    if e == -1:
        e = len(chosen)
    return chosen[s:e]


def prepare_hh_prompts(records: Iterable[dict], tokenizer,
                       max_prompt_len: Optional[int] = None,
                       template: str = QWEN_CHAT_TEMPLATE) -> list[list[int]]:
    """hh-rlhf-style records ({'chosen': transcript}) → tokenized prompts
    wrapped in the chat scaffold (grpo.py:249-270)."""
    prompts = []
    for rec in records:
        question = extract_hh_question(rec["chosen"]) if "chosen" in rec \
            else rec["question"]
        ids = tokenizer(template.replace("QUESTION", question),
                        padding=False)["input_ids"]
        if max_prompt_len is not None and len(ids) > max_prompt_len:
            continue  # reference filters long prompts rather than truncating
        prompts.append(ids)
    return prompts


MATH_TEMPLATE = (
    "<|im_start|>system\nYou are a helpful assistant. Solve the problem and "
    "put your final answer within \\boxed{}.<|im_end|>\n"
    "<|im_start|>user\nQUESTION<|im_end|>\n<|im_start|>assistant\n"
)


def prepare_math_prompts(records: Iterable[dict], tokenizer,
                         max_prompt_len: Optional[int] = None,
                         template: str = MATH_TEMPLATE,
                         ) -> tuple[list[list[int]], dict[str, str]]:
    """r1-mode (question, answer) records → (tokenized prompts,
    prompt-text → gold-answer map), mirroring grpo_r1.py:237-246's
    train-set answer hash map keyed by the prompt string."""
    prompts, answers = [], {}
    for rec in records:
        q = rec["question"] if "question" in rec else rec["query"]
        text = template.replace("QUESTION", q)
        ids = tokenizer(text, padding=False)["input_ids"]
        if max_prompt_len is not None and len(ids) > max_prompt_len:
            continue
        prompts.append(ids)
        answers[text] = str(rec.get("answer", ""))
    return prompts, answers


def make_tiny_tokenizer(save_dir: str, vocab_size: int = 1024):
    """Build + save a tiny self-contained byte-level BPE tokenizer for tests
    and offline demos (no network: real checkpoints aren't downloadable in
    this environment, so the tokenizer seam is exercised with this)."""
    import os

    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers
    from transformers import PreTrainedTokenizerFast

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=vocab_size,
        special_tokens=["<|im_start|>", "<|im_end|>", "<|endoftext|>"],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet())
    corpus = ["Human: hello there Assistant: hi", "What is 2+2? \\boxed{4}",
              "You are Qwen, created by Alibaba Cloud.",
              "You are a helpful assistant. Solve the problem."]
    tok.train_from_iterator(corpus * 50, trainer)
    fast = PreTrainedTokenizerFast(tokenizer_object=tok,
                                   eos_token="<|im_end|>",
                                   bos_token=None, unk_token=None)
    os.makedirs(save_dir, exist_ok=True)
    fast.save_pretrained(save_dir)
    return save_dir
