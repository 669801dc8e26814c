#!/usr/bin/env python3
"""Operator check: will this HuggingFace checkpoint dir serve?

    python tools/hf_check.py /models/Qwen3-8B [--load]

Prints the parsed config (family detection, rope scaling, head_dim,
tokenizer presence) and, with --load, actually loads the weights on CPU
and runs one tiny forward so name-mapping/shape problems surface here
instead of at instance-create time.
"""
import argparse
import os
import sys

sys.path.insert(0, ".")

import torch  # noqa: E402

from fma_amd.models import loader  # noqa: E402
from fma_amd.models.llama import LlamaModel  # noqa: E402


def main():
    ap = argparse.ArgumentParser("fma-hf-check")
    ap.add_argument("path")
    ap.add_argument("--load", action="store_true",
                    help="load weights on CPU and run one forward")
    args = ap.parse_args()
    cfg = loader.config_from_dir(args.path)
    if cfg is None:
        print("no config.json found", file=sys.stderr)
        return 2
    gib = cfg.total_param_bytes() / 2**30
    print(f"model        {cfg.name}")
    print(f"shape        hidden={cfg.hidden_size} layers={cfg.num_layers} "
          f"heads={cfg.num_heads}/{cfg.num_kv_heads} hd={cfg.head_dim} "
          f"vocab={cfg.vocab_size}")
    feats = [k for k, v in (
        ("qkv_bias", cfg.qkv_bias), ("qk_norm", cfg.qk_norm),
        ("embed_scale", cfg.embed_scale),
        ("gelu", cfg.hidden_act != "silu"),
        (f"moe({cfg.num_experts} experts, "
         f"top-{cfg.num_experts_per_tok})", bool(cfg.num_experts)),
        ("rope_scaling", bool(cfg.rope_scaling))) if v]
    print(f"features     {', '.join(feats) or 'plain llama'}")
    print(f"params       {gib:.2f} GiB @ {cfg.dtype}")
    print(f"tokenizer    "
          f"{'tokenizer.json' if os.path.exists(os.path.join(args.path, 'tokenizer.json')) else 'NONE (byte-level fallback)'}")
    nshards = len(loader.shard_files(args.path))
    print(f"safetensors  {nshards} shard(s)")
    if not args.load:
        return 0
    params = {n: torch.zeros(s, dtype=d)
              for n, s, d in cfg.param_specs()}
    n = loader.load_into_params(args.path, params, cfg=cfg)
    model = LlamaModel(cfg, params, torch.device("cpu"))
    toks = torch.randint(0, cfg.vocab_size, (1, 4))
    logits = model.forward(toks)
    print(f"load OK      {n} tensors; forward logits "
          f"{tuple(logits.shape)}, finite="
          f"{bool(torch.isfinite(logits).all())}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
