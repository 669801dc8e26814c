import time

import torch

from fma_amd.models.llama import LlamaConfig
from fma_amd.runtime.engine import ActuationEngine

cfg = LlamaConfig.by_name("mixtral-8x7b")
cfg.max_seq_len = 2048
t0 = time.perf_counter()
eng = ActuationEngine(cfg, seed=3)
print(f"mixtral-8x7b {eng.total_bytes/2**30:.1f} GiB up in "
      f"{time.perf_counter()-t0:.1f}s", flush=True)
toks = torch.randint(0, cfg.vocab_size, (1, 64), device=eng.device)
eng.generate(toks, max_new_tokens=8)
torch.cuda.synchronize()
t0 = time.perf_counter()
eng.generate(toks, max_new_tokens=24)
torch.cuda.synchronize()
tps = 24 / (time.perf_counter() - t0)
ts = eng.sleep()
tw = eng.wake_up()
print(f"decode {tps:.1f} tok/s; sleep {ts:.2f}s wake {tw:.2f}s", flush=True)
