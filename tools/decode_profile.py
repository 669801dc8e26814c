"""Focused decode-at-context profile target for rocprofv3 --stats."""
import sys
import time

import torch

from fma_amd.models.llama import LlamaConfig
from fma_amd.runtime.engine import ActuationEngine

ctx = int(sys.argv[1]) if len(sys.argv) > 1 else 4000
new = int(sys.argv[2]) if len(sys.argv) > 2 else 64
cfg = LlamaConfig.by_name("synthetic-15gib")
eng = ActuationEngine(cfg, seed=7)
toks = torch.randint(0, cfg.vocab_size, (1, ctx), device=eng.device)
eng.generate(toks, max_new_tokens=4)  # warm + prefill path
torch.cuda.synchronize()
t0 = time.perf_counter()
eng.generate(toks, max_new_tokens=new)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"ctx={ctx} decode {new} tokens (plus prefill) in {dt:.3f}s")
