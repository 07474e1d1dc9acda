"""Probe: does the prefill hipGraph capture succeed with capture-safe GEMMs?

Runs the tiny model through LocalEngine.generate with a long prompt so the
prefill-graph path triggers, then asserts the graph actually captured (not
the eager fallback) and the output is sane. Run under its own `timeout` so
a capture hang can't eat the box.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("ROOMAMD_MODEL_CONFIG", "tiny")

import torch  # noqa: E402

from room_amd.engine.llm import LocalEngine  # noqa: E402

eng = LocalEngine()
prompt = list(range(1, 1200))  # > 1 bucket boundary (512, 1024)
req = eng.generate(prompt, max_new_tokens=16, session_key="probe", timeout=120)
assert len(req.out_tokens) == 16, req.out_tokens
print("prefill_graphs captured:", sorted(eng._prefill_graphs.keys()))
print("prefill_graphs_broken:", eng._prefill_graphs_broken)
print("stats:", {k: round(v, 4) if isinstance(v, float) else v
                 for k, v in eng.stats.items()})
if eng._prefill_graphs_broken:
    print("PROBE: CAPTURE FAILED (eager fallback ran)")
    sys.exit(1)
if not eng._prefill_graphs:
    print("PROBE: graph path never triggered")
    sys.exit(2)
# numerics: same prompt through a fresh engine with graphs off must produce
# the same greedy-ish trajectory shape (sampling is seeded per-slot, so just
# check the graph run produced in-vocab tokens)
V = eng.cfg.vocab_size
assert all(0 <= t < V for t in req.out_tokens), req.out_tokens
# second generate reusing the session (prefix reuse + small tail chunk)
req2 = eng.generate(prompt + req.out_tokens + list(range(5, 600)),
                    max_new_tokens=8, session_key="probe", timeout=120)
assert len(req2.out_tokens) == 8
assert not eng._prefill_graphs_broken
print("PROBE: OK")
eng.shutdown()
