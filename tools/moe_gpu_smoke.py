"""MoE on MI355X: decode + sleep/wake with a mid-size synthetic MoE."""
import time

import torch

from fma_amd.models.llama import LlamaConfig
from fma_amd.runtime.engine import ActuationEngine

# tiny first: bit-stability
cfg = LlamaConfig.by_name("tiny-moe")
eng = ActuationEngine(cfg, seed=4)
toks = torch.randint(0, cfg.vocab_size, (1, 6), device=eng.device)
o1 = eng.generate(toks, max_new_tokens=5).clone()
eng.sleep(); eng.wake_up()
assert torch.equal(eng.generate(toks, max_new_tokens=5), o1)
print("tiny-moe OK", flush=True)
del eng
torch.cuda.empty_cache()

# mid-size: 8 experts x (4096->14336) x 8 layers ~ 16.5 GiB
cfg = LlamaConfig(name="moe-16g", vocab_size=32768, hidden_size=4096,
                  intermediate_size=14336, num_layers=8, num_heads=32,
                  num_kv_heads=8, max_seq_len=4096, num_experts=8,
                  num_experts_per_tok=2)
t0 = time.perf_counter()
eng = ActuationEngine(cfg, seed=6)
print(f"engine {eng.total_bytes/2**30:.1f} GiB up in "
      f"{time.perf_counter()-t0:.1f}s", flush=True)
toks = torch.randint(0, cfg.vocab_size, (1, 64), device=eng.device)
out = eng.generate(toks, max_new_tokens=16)
torch.cuda.synchronize()
t0 = time.perf_counter()
out = eng.generate(toks, max_new_tokens=32)
torch.cuda.synchronize()
tps = 32 / (time.perf_counter() - t0)
ts = eng.sleep()
tw = eng.wake_up()
out2 = eng.generate(toks, max_new_tokens=32)
assert torch.equal(out, out2), "post-wake divergence"
print(f"moe-16g decode {tps:.1f} tok/s (incl prefill amort), "
      f"sleep {ts:.2f}s wake {tw:.2f}s, post-wake bit-stable", flush=True)
