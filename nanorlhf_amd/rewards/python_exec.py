"""Sandboxed Python executor (parity with the reference's vendored
examples/r1-v0/utils/eval/python_executor.py:42-135 — unused by its
training path, provided here for tool-augmented reward functions).

Runs a code snippet in a subprocess with a hard wall-clock timeout, no
network assumptions, captured stdout, and an optional `answer` variable
convention."""
from __future__ import annotations

import multiprocessing as mp


def _worker(code: str, q):  # pragma: no cover - subprocess body
    import contextlib
    import io
    buf = io.StringIO()
    env: dict = {"__name__": "__main__"}
    try:
        with contextlib.redirect_stdout(buf):
            exec(code, env)  # noqa: S102 - sandboxed by subprocess + timeout
        q.put({"ok": True, "stdout": buf.getvalue(),
               "answer": env.get("answer")})
    except Exception as e:  # noqa: BLE001
        q.put({"ok": False, "stdout": buf.getvalue(), "error": repr(e)})


class PythonExecutor:
    def __init__(self, timeout_s: float = 5.0):
        self.timeout_s = timeout_s

    def run(self, code: str) -> dict:
        ctx = mp.get_context("fork")
        q = ctx.Queue()
        p = ctx.Process(target=_worker, args=(code, q), daemon=True)
        p.start()
        p.join(self.timeout_s)
        if p.is_alive():
            p.terminate()
            p.join(0.2)
            return {"ok": False, "stdout": "", "error": "timeout"}
        try:
            return q.get_nowait()
        except Exception:  # noqa: BLE001
            return {"ok": False, "stdout": "", "error": "no result"}
