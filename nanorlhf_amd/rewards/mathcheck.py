"""Math-answer equivalence checking (own implementation).

Covers the role of the reference's vendored verifier stack
(examples/r1-v0/utils/toolkit_for_MATH/latex_answer_check.py:52-236,
parsing_lib.py:71-230, metamath_utils.py:171-253 and
utils/eval/eval_utils.py:181-278): extract \\boxed{...} answers, normalize
LaTeX, then compare STRUCTURALLY —

  * intervals incl. \\cup unions, with bracket-type (open/closed) semantics
  * ordered tuples (…)/[…] and unordered sets {…}
  * matrices (pmatrix/bmatrix/vmatrix/matrix environments), elementwise
  * percent, units, degrees, mixed numbers (2\\frac{1}{2}), scientific
    notation, thousands separators, equation RHS extraction (x = 5 → 5)

falling through to a 3-stage scalar check: literal (whitespace/comma
insensitive) → numeric (int exact, float rel 1e-3 — the reference's
tolerance) → sympy symbolic difference.  The sympy stage runs under a
subprocess timeout (reference call_with_timeout, grpo_r1.py:179-224)
because sympy can hang on adversarial inputs.
"""
from __future__ import annotations

import multiprocessing as mp
import re
from dataclasses import dataclass


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


# ---------------------------------------------------------------------------
# normalization (lexical; keeps bracket structure for the structured parser)
# ---------------------------------------------------------------------------

_UNIT_WORDS = (
    "degrees|degree|units|unit|cm|mm|km|m|ft|feet|foot|inches|inch|in|yards|"
    "yard|miles|mile|meters|meter|seconds|second|sec|s|minutes|minute|min|"
    "hours|hour|h|days|day|years|year|mph|dollars|dollar|cents|cent|pounds|"
    "pound|kg|g|grams|gram|liters|liter|L|square +(?:units|cm|m|km|inches|feet|miles)|"
    "cubic +(?:units|cm|m|km|inches|feet)|cm\\^2|m\\^2|cm\\^3|m\\^3|cm\\*\\*2|m\\*\\*2"
)

_LATEX_SUBS = [
    (r"\\left\s*\.", ""), (r"\\right\s*\.", ""),
    (r"\\left", ""), (r"\\right", ""), (r"\\!", ""), (r"\\,", ""), (r"\\;", ""),
    (r"\\:", ""), (r"~", " "),
    (r"\\ ", " "), (r"\\\$", ""), (r"\$", ""),
    (r"\\text\s*\{([^}]*)\}", r" \1"), (r"\\textbf\s*\{([^}]*)\}", r" \1"),
    (r"\\mathrm\s*\{([^}]*)\}", r" \1"), (r"\\mbox\s*\{([^}]*)\}", r" \1"),
    (r"\\mathbf\s*\{([^}]*)\}", r"\1"), (r"\\mathbb\s*\{([^}]*)\}", r"\1"),
    (r"\\operatorname\s*\{([^}]*)\}", r"\1"),
    (r"\\dfrac", r"\\frac"), (r"\\tfrac", r"\\frac"), (r"\\cfrac", r"\\frac"),
    (r"\\cdot", "*"), (r"\\times", "*"), (r"\\div", "/"),
    (r"\\pi", "pi"), (r"\\infty", "oo"), (r"\\infinity", "oo"), (r"∞", "oo"),
    (r"°", ""), (r"\^\s*\\circ", ""), (r"\\circ", ""),
    (r"\\%", "%"), (r"%", "%"),
    (r"\\\{", "{"), (r"\\\}", "}"),
    (r"\\le(?![a-z])", "<="), (r"\\leq", "<="), (r"\\ge(?![a-z])", ">="),
    (r"\\geq", ">="), (r"\\neq", "!="), (r"\\ne(?![a-z])", "!="),
]


def _convert_fracs(s: str) -> str:
    # mixed numbers FIRST: 2\frac{1}{2} → (2+(1)/(2)); 2 1/2 likewise
    s = re.sub(r"(?<![\d.a-zA-Z)])(\d+)\s*\\frac\{(\d+)\}\{(\d+)\}",
               r"(\1+(\2)/(\3))", s)
    s = re.sub(r"(?<![\d.a-zA-Z)/])(\d+)\s+(\d+)\s*/\s*(\d+)(?![\d.])",
               r"(\1+(\2)/(\3))", s)
    # frac/sqrt to fixpoint (innermost-out; mixed nesting like
    # \frac{\sqrt{2}}{2} needs the two rules to alternate)
    s = re.sub(r"\\sqrt\s*\[(\d+)\]\s*\{([^{}]*)\}", r"((\2)**(1/\1))", s)
    for _ in range(8):
        s2 = re.sub(r"\\frac\s*\{([^{}]*)\}\s*\{([^{}]*)\}", r"((\1)/(\2))", s)
        s2 = re.sub(r"\\sqrt\s*\{([^{}]*)\}", r"sqrt(\1)", s2)
        if s2 == s:
            break
        s = s2
    s = re.sub(r"\\frac\s*(\d)\s*(\d)", r"((\1)/(\2))", s)
    s = re.sub(r"\\frac\s*\{([^{}]*)\}\s*(\d)", r"((\1)/(\2))", s)
    s = re.sub(r"\\sqrt\s*(\d)", r"sqrt(\1)", s)
    return s


def normalize_latex(ans: str) -> str:
    """LaTeX → plain-ish, KEEPING (), [], {} bracket structure
    (parsing_lib.py string_normalization role)."""
    s = ans.strip()
    s = re.sub(r"\\boxed\s*\{(.*)\}\s*$", r"\1", s)  # nested boxes: keep content
    for pat, rep in _LATEX_SUBS:
        s = re.sub(pat, rep, s)
    s = _convert_fracs(s)
    # thousands separators inside numbers: 1,234,567 (NOT tuple commas —
    # require 3-digit groups flanked by digits)
    s = re.sub(r"(?<=\d),(?=\d{3}(\D|$))", "", s)
    s = re.sub(r"\s+", " ", s)
    return s.strip().strip(".").strip()


def _strip_units(s: str) -> str:
    """Trailing unit words after a numeric/symbolic answer: '5 cm' → '5'."""
    return re.sub(r"^(.*?\S)\s+(" + _UNIT_WORDS + r")\.?\s*$", r"\1", s)


def _strip_equation(s: str) -> str:
    """'x = 5' → '5' (reference remove_equals, parsing_lib.py:230-240):
    single '=' with nonempty sides keeps the RHS."""
    if s.count("=") == 1 and "<=" not in s and ">=" not in s and "!=" not in s:
        lhs, rhs = s.split("=")
        if lhs.strip() and rhs.strip():
            return rhs.strip()
    return s


def normalize_answer(ans: str) -> str:
    """Full scalar normalization to an expression string (sympy-ready)."""
    s = normalize_latex(ans)
    s = _strip_equation(s)
    s = _strip_units(s)
    s = s.replace("^", "**").replace("{", "(").replace("}", ")")
    s = s.strip(" .$")
    s = s.lstrip("+").strip()
    return s


# ---------------------------------------------------------------------------
# structured answers: intervals/tuples/sets/matrices
# ---------------------------------------------------------------------------

@dataclass
class Structured:
    kind: str            # "interval_union" | "tuple" | "set" | "matrix"
    parts: list          # elements; for interval_union: (lb, items, rb) triples


def _split_top_level(s: str, sep: str = ",") -> list[str]:
    out, depth, cur = [], 0, []
    for ch in s:
        if ch in "([{":
            depth += 1
        elif ch in ")]}":
            depth -= 1
        if ch == sep and depth == 0:
            out.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    out.append("".join(cur))
    return [x.strip() for x in out]


_MATRIX_RE = re.compile(
    r"\\begin\s*\{(?:p|b|v|B|small)?matrix\}(.*?)\\end\s*\{(?:p|b|v|B|small)?matrix\}",
    re.S)


def parse_structured(ans: str) -> Structured | None:
    """Detect interval unions, tuples, sets, matrices in a normalized-latex
    string; None → plain scalar."""
    s = normalize_latex(ans).strip()
    m = _MATRIX_RE.search(s)
    if m:
        rows = [r for r in re.split(r"\\\\", m.group(1)) if r.strip()]
        parts = [[c.strip() for c in row.split("&")] for row in rows]
        return Structured("matrix", parts)
    s = _strip_equation(s)
    # interval union:  (a,b] \cup [c, oo)
    pieces = re.split(r"\\cup|∪", s)
    iv = []
    for p in pieces:
        p = p.strip()
        if len(p) >= 2 and p[0] in "[(" and p[-1] in "])":
            elems = _split_top_level(p[1:-1])
            if len(elems) == 2 and all(elems):
                iv.append((p[0], elems, p[-1]))
                continue
        iv = None
        break
    if iv is not None:
        # a bare (a,b)/[a,b] is an interval OR an ordered pair — either way
        # the comparison (same brackets + elementwise) is identical, so one
        # representation serves both
        return Structured("interval_union", iv)
    # tuple / set / longer ordered list
    m3 = re.fullmatch(r"([\[({])\s*(.*?)\s*([\])}])", s, re.S)
    if m3 and "," in m3.group(2):
        elems = _split_top_level(m3.group(2))
        if m3.group(1) == "{" and m3.group(3) == "}":
            return Structured("set", elems)
        return Structured("tuple", [m3.group(1), elems, m3.group(3)])
    return None


# ---------------------------------------------------------------------------
# scalar equivalence (literal → numeric → sympy)
# ---------------------------------------------------------------------------

def _num(s: str) -> float | None:
    s = s.strip()
    try:
        if s.endswith("%"):
            return float(s[:-1]) / 100.0
        return float(s)
    except ValueError:
        pass
    # scientific latex: 1.2*10**5
    m = re.fullmatch(r"(-?\d+(?:\.\d+)?)\s*\*\s*10\s*\*\*\s*\(?(-?\d+)\)?", s)
    if m:
        return float(m.group(1)) * 10.0 ** int(m.group(2))
    return None


def _sympy_equal(a: str, b: str) -> bool:
    import sympy
    from sympy.parsing.sympy_parser import (implicit_multiplication_application,
                                            parse_expr, standard_transformations)
    tr = standard_transformations + (implicit_multiplication_application,)
    ea = parse_expr(a, transformations=tr, evaluate=True)
    eb = parse_expr(b, transformations=tr, evaluate=True)
    try:
        if sympy.simplify(ea - eb) == 0:
            return True
    except Exception:  # noqa: BLE001
        pass
    try:
        na, nb = complex(sympy.N(ea)), complex(sympy.N(eb))
        return abs(na - nb) <= 1e-3 * max(1.0, abs(nb))
    except Exception:  # noqa: BLE001
        return False


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


def _scalar_equal(pred: str, gold: str, sympy_timeout_s: float) -> bool:
    a, b = normalize_answer(pred), normalize_answer(gold)
    if not a or not b:
        return False
    # literal, whitespace-insensitive (latex_answer_check literal_check role)
    if a.replace(" ", "") == b.replace(" ", ""):
        return True
    na, nb = _num(a), _num(b)
    if na is not None and nb is not None:
        # int-exact / float rel 1e-3 (reference numerical_equal semantics)
        if float(na).is_integer() and float(nb).is_integer() and na == nb:
            return True
        if abs(na - nb) <= 1e-3 * max(1.0, abs(nb)):
            return True
    # percent laxness: '50%' ≡ '50' as well as '0.5'
    if a.endswith("%") != b.endswith("%"):
        xa, xb = _num(a.rstrip("%")), _num(b.rstrip("%"))
        if xa is not None and xb is not None:
            return abs(xa - xb) <= 1e-3 * max(1.0, abs(xb))
    if na is not None and nb is not None:
        return False
    if re.fullmatch(r"-?[\d. ]+%?", a) and re.fullmatch(r"-?[\d. ]+%?", b):
        return False  # both plain numbers and not close — don't burn a subprocess
    # warm sympy in the PARENT: the timeout-guarded child is a fork and must
    # not pay the multi-second sympy import inside its budget
    import sympy  # noqa: F401
    import sympy.parsing.sympy_parser  # noqa: F401
    return call_with_timeout(_sympy_worker, (a.rstrip("%"), b.rstrip("%")),
                             sympy_timeout_s)


def _structured_equal(sa: Structured, sb: Structured, timeout_s: float) -> bool:
    if sa.kind != sb.kind:
        # (a,b) can be an interval OR a pair — both encode as interval_union
        # vs tuple only when lengths differ → not equal
        return False
    if sa.kind == "matrix":
        if len(sa.parts) != len(sb.parts):
            return False
        for ra, rb in zip(sa.parts, sb.parts):
            if len(ra) != len(rb):
                return False
            if not all(_scalar_equal(x, y, timeout_s) for x, y in zip(ra, rb)):
                return False
        return True
    if sa.kind == "interval_union":
        if len(sa.parts) != len(sb.parts):
            return False
        for (lb_a, it_a, rb_a), (lb_b, it_b, rb_b) in zip(sa.parts, sb.parts):
            if lb_a != lb_b or rb_a != rb_b:  # open/closed semantics differ
                return False
            if not all(_scalar_equal(x, y, timeout_s) for x, y in zip(it_a, it_b)):
                return False
        return True
    if sa.kind == "tuple":
        lb_a, el_a, rb_a = sa.parts
        lb_b, el_b, rb_b = sb.parts
        if len(el_a) != len(el_b):
            return False
        return all(_scalar_equal(x, y, timeout_s) for x, y in zip(el_a, el_b))
    if sa.kind == "set":
        if len(sa.parts) != len(sb.parts):
            return False
        used = [False] * len(sb.parts)
        for x in sa.parts:
            hit = False
            for j, y in enumerate(sb.parts):
                if not used[j] and _scalar_equal(x, y, timeout_s):
                    used[j] = True
                    hit = True
                    break
            if not hit:
                return False
        return True
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
    """Structured-first equivalence (latex_answer_check.py:166-236 role,
    widened to the vendored stack's interval/tuple/set/matrix breadth)."""
    if pred is None or gold is None:
        return False
    sa, sb = parse_structured(pred), parse_structured(gold)
    if sa is not None or sb is not None:
        # tuple-vs-interval ambiguity: a bare (a,b)/[a,b] parses as
        # interval_union; compare whatever both parse to
        if sa is None or sb is None:
            return False
        return _structured_equal(sa, sb, sympy_timeout_s)
    return _scalar_equal(pred, gold, sympy_timeout_s)


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
