"""Math-answer equivalence checking (own implementation).

Covers the role of the reference's vendored verifier stack
(examples/r1-v0/utils/toolkit_for_MATH/latex_answer_check.py and
utils/eval/eval_script.py): extract \\boxed{...} answers, normalize LaTeX,
then test equivalence by (1) literal match, (2) numeric compare,
(3) sympy symbolic difference — the sympy stage runs under a subprocess
timeout (reference call_with_timeout, grpo_r1.py:179-224) because sympy can
hang on adversarial inputs."""
from __future__ import annotations

import multiprocessing as mp
import re


def extract_boxed(text: str) -> str | None:
    """Last \\boxed{...} with balanced braces (grpo_r1.py:194-213 get_boxed role)."""
    idx = text.rfind("\\boxed")
    if idx < 0:
        return None
    i = idx + len("\\boxed")
    while i < len(text) and text[i] in " \t":
        i += 1
    if i >= len(text):
        return None
    if text[i] != "{":
        # \boxed 42 style
        m = re.match(r"([^$\\}\s]+)", text[i:])
        return m.group(1) if m else None
    depth = 0
    start = i + 1
    for j in range(i, len(text)):
        if text[j] == "{":
            depth += 1
        elif text[j] == "}":
            depth -= 1
            if depth == 0:
                return text[start:j]
    return None


_LATEX_SUBS = [
    (r"\\left", ""), (r"\\right", ""), (r"\\!", ""), (r"\\,", ""), (r"\\;", ""),
    (r"\\ ", " "), (r"\\%", "%"), (r"\\\$", ""), (r"\$", ""), (r"\\text\{([^}]*)\}", r"\1"),
    (r"\\mathrm\{([^}]*)\}", r"\1"), (r"\\mbox\{([^}]*)\}", r"\1"),
    (r"\\dfrac", r"\\frac"), (r"\\tfrac", r"\\frac"),
    (r"\\cdot", "*"), (r"\\times", "*"), (r"\\div", "/"),
    (r"\\pi", "pi"), (r"\\infty", "oo"), (r"°", ""), (r"\\circ", ""),
    (r"\s+", " "),
]


def normalize_answer(ans: str) -> str:
    """LaTeX → plain-ish normalization (parsing_lib.py:71-230 role)."""
    s = ans.strip()
    for pat, rep in _LATEX_SUBS:
        s = re.sub(pat, rep, s)
    # \frac{a}{b} -> (a)/(b)
    for _ in range(4):
        s2 = re.sub(r"\\frac\{([^{}]*)\}\{([^{}]*)\}", r"((\1)/(\2))", s)
        if s2 == s:
            break
        s = s2
    s = re.sub(r"\\frac(\d)(\d)", r"((\1)/(\2))", s)
    s = re.sub(r"\\sqrt\{([^{}]*)\}", r"sqrt(\1)", s)
    s = re.sub(r"\\sqrt(\d)", r"sqrt(\1)", s)
    s = s.replace("^", "**").replace("{", "(").replace("}", ")")
    s = s.strip(" .$")
    # strip trailing units-ish words
    s = re.sub(r"\s*(degrees|units|cm|mm|m|ft|inches|in)\.?$", "", s)
    # drop thousands separators in pure numbers: 1,234,567
    if re.fullmatch(r"-?\d{1,3}(,\d{3})+(\.\d+)?", s):
        s = s.replace(",", "")
    return s.strip()


def _num(s: str) -> float | None:
    try:
        if s.endswith("%"):
            return float(s[:-1]) / 100.0
        return float(s)
    except ValueError:
        return None


def _sympy_equal(a: str, b: str) -> bool:
    import sympy
    from sympy.parsing.sympy_parser import parse_expr, standard_transformations, \
        implicit_multiplication_application
    tr = standard_transformations + (implicit_multiplication_application,)
    ea = parse_expr(a.replace("**", "^").replace("^", "**"), transformations=tr, evaluate=True)
    eb = parse_expr(b.replace("**", "^").replace("^", "**"), transformations=tr, evaluate=True)
    diff = sympy.simplify(ea - eb)
    return diff == 0


def _sympy_worker(a, b, q):  # pragma: no cover - subprocess body
    try:
        q.put(bool(_sympy_equal(a, b)))
    except Exception:  # noqa: BLE001
        q.put(False)


def call_with_timeout(fn, args, timeout_s: float) -> bool:
    """Run fn(*args, queue) in a subprocess with a hard timeout
    (reference grpo_r1.py:179-192)."""
    ctx = mp.get_context("fork")
    q = ctx.Queue()
    p = ctx.Process(target=fn, args=(*args, q), daemon=True)
    p.start()
    p.join(timeout_s)
    if p.is_alive():
        p.terminate()
        p.join(0.2)
        return False
    try:
        return bool(q.get_nowait())
    except Exception:  # noqa: BLE001
        return False


def extract_math_answer(text: str) -> str | None:
    """Boxed answer if present, else the last number in the text (the
    reference's answer-extraction strategies, utils/data_processing/
    answer_extraction.py:177-245 role)."""
    pred = extract_boxed(text)
    if pred is not None:
        return pred
    nums = re.findall(r"-?\d+(?:\.\d+)?", text)
    return nums[-1] if nums else None


def answers_equal(pred: str, gold: str, sympy_timeout_s: float = 0.5) -> bool:
    """3-stage equivalence (latex_answer_check.py:166-236 role)."""
    a, b = normalize_answer(pred), normalize_answer(gold)
    if not a or not b:
        return False
    if a == b:
        return True
    na, nb = _num(a), _num(b)
    if na is not None and nb is not None:
        return abs(na - nb) <= 1e-6 * max(1.0, abs(nb))
    # warm sympy in the PARENT: the timeout-guarded child is a fork and must
    # not pay the multi-second sympy import inside its budget
    import sympy  # noqa: F401
    import sympy.parsing.sympy_parser  # noqa: F401
    return call_with_timeout(_sympy_worker, (a, b), sympy_timeout_s)


# API-parity aliases for the reference's verifier entry points
# (latex_answer_check.py:166 latex_answer_check; eval_script.py:6 is_correct;
# eval_utils.py:181 math_equal; metamath_utils.py:171 is_equiv)
def latex_answer_check(pred: str, gold: str, timeout_s: float = 0.5) -> bool:
    return answers_equal(pred, gold, timeout_s)


def is_correct(pred: str, gold: str, timeout_s: float = 0.5) -> bool:
    return answers_equal(pred, gold, timeout_s)


def math_equal(pred: str, gold: str, timeout_s: float = 0.5) -> bool:
    return answers_equal(pred, gold, timeout_s)


def is_equiv(pred: str, gold: str, timeout_s: float = 0.5) -> bool:
    return answers_equal(pred, gold, timeout_s)
