"""Tiny decode workload for PMC collection: B=5 agents, short generates."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import threading

from room_amd.engine import tokenizer as tok
from room_amd.engine.llm import LocalEngine

eng = LocalEngine()
prompts = [tok.encode(f"agent {i} ctx " * 400) for i in range(5)]  # ~1.2k tok

def run(i):
    eng.generate(prompts[i], max_new_tokens=8, session_key=f"p{i}", timeout=300)

threads = [threading.Thread(target=run, args=(i,)) for i in range(5)]
for t in threads:
    t.start()
for t in threads:
    t.join()
print("decode pmc workload done", flush=True)
eng.shutdown()
