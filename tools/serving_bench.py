"""Serving-path latency: T_first_token / T_e2e (reference benchmark.md
metric definitions the reference never measures).

Starts the serving runtime in-process, measures time-to-first-token for a
completion right after wake (the worst case a router sees after an
actuation) and in steady state.

GPU:  python tools/serving_bench.py --model synthetic-15gib
CPU:  works with --model tiny (fake arena).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from fma_amd.runtime.server import ServingRuntime, parse_options  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="tiny")
    ap.add_argument("--prompt-len", type=int, default=64)
    ap.add_argument("--new-tokens", type=int, default=32)
    ap.add_argument("--reps", type=int, default=3)
    args = ap.parse_args()

    rt = ServingRuntime(parse_options(f"--model {args.model}"))
    r = rt.rt
    eng0 = r.engine if hasattr(r, "engine") else r
    toks = torch.randint(0, eng0.cfg.vocab_size, (1, args.prompt_len),
                         device=eng0.device)

    def first_token_seconds():
        t0 = time.perf_counter()
        eng = r.engine if hasattr(r, "engine") else r
        cache = eng.new_kv_cache(1, args.prompt_len + 2)
        eng.model.forward(toks.to(eng.device), cache, 0)
        if eng.on_gpu:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        cache.free()
        return dt

    def decode_tps():
        """End-to-end rate of generate() (prefill + decode)."""
        eng = r.engine if hasattr(r, "engine") else r
        t0 = time.perf_counter()
        eng.generate(toks, max_new_tokens=args.new_tokens)
        if eng.on_gpu:
            torch.cuda.synchronize()
        return args.new_tokens / (time.perf_counter() - t0)

    def pure_decode_tps():
        """Steady-state single-token rate: prefill once, then time only
        the decode steps against the warm KV cache."""
        eng = r.engine if hasattr(r, "engine") else r
        cache = eng.new_kv_cache(1, args.prompt_len + args.new_tokens + 2)
        logits = eng.model.forward(toks.to(eng.device), cache, 0)
        nxt = logits[:, -1:].argmax(-1)
        if eng.on_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        pos = args.prompt_len
        for _ in range(args.new_tokens):
            logits = eng.model.forward(nxt, cache, pos)
            nxt = logits[:, -1:].argmax(-1)
            pos += 1
        if eng.on_gpu:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        cache.free()
        return args.new_tokens / dt

    # steady state
    first_token_seconds()
    steady_ttft = min(first_token_seconds() for _ in range(args.reps))
    steady_tps = max(decode_tps() for _ in range(args.reps))
    steady_pure_tps = max(pure_decode_tps() for _ in range(args.reps))

    # right after a wake (allocator cold, caches dropped)
    post_wake_ttft = []
    for _ in range(args.reps):
        r.sleep(1)
        r.wake_up()
        post_wake_ttft.append(first_token_seconds())

    graphed_tps = None
    eng = r.engine if hasattr(r, "engine") else r
    from fma_amd.models.decode_graph import StaticDecoder
    if eng.on_gpu and eng.tp_size == 1 \
            and StaticDecoder.supported(eng.cfg):
        dec = StaticDecoder(eng.model, 1, args.prompt_len + args.new_tokens + 2)
        dec.capture()
        dec.generate(toks, args.new_tokens)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        dec.generate(toks, args.new_tokens)
        torch.cuda.synchronize()
        graphed_tps = args.new_tokens / (time.perf_counter() - t0)

    print(json.dumps({
        "metric": "T_first_token / decode throughput",
        "model": args.model,
        "prompt_len": args.prompt_len,
        "steady_ttft_s": round(steady_ttft, 4),
        "post_wake_ttft_s": round(min(post_wake_ttft), 4),
        "decode_tok_s": round(steady_tps, 2),
        "decode_tok_s_pure": round(steady_pure_tps, 2),
        "decode_tok_s_hipgraph": round(graphed_tps, 2) if graphed_tps else None,
    }))
    if hasattr(r, "stop"):
        r.stop()


if __name__ == "__main__":
    main()
