"""Actuation engine: a model instance with fast sleep(level=1)/wake_up.

This is the MI355X-native replacement for the slice of vLLM the reference
depends on (sleep mode + a serving endpoint; reference README.md:16-33,
docs/dual-pods.md:618-627). One engine == one model instance on one GPU
(or one TP rank of it):

- parameters are views into a contiguous DeviceArena (ops/actuation.py), so
  ``sleep`` is chunked pinned D2H + physical-HBM release and ``wake_up`` is
  physical re-map overlapped with pinned H2D — no allocator round-trips, no
  module re-init on the hot path;
- the pinned host buffer is allocated once at instance creation (the
  launcher's create step), keeping hipHostMalloc cost off the wake path;
- for TP instances every rank restores its own shard locally (no inter-GPU
  weight traffic over xGMI) and an RCCL barrier gates the transition to
  ``is_sleeping == False`` so all ranks re-enter serving together.
"""

from __future__ import annotations

import os
import time
from typing import Dict, Optional

import torch
import torch.distributed as dist

from fma_amd.models.llama import KVCache, LlamaConfig, LlamaModel
from fma_amd.ops import actuation


class ActuationEngine:
    AWAKE = "awake"
    SLEEPING = "sleeping"

    def __init__(self, cfg: LlamaConfig, device_index: int = 0,
                 tp_rank: int = 0, tp_size: int = 1, tp_group=None,
                 use_vmm=None, chunk_bytes: int = 0, seed: int = 0,
                 init_weights: bool = True, nstreams: int = 1,
                 slab_bytes: int = None, actuation_mode: str = "arena",
                 pack_xfer_mode=None):
        self.cfg = cfg
        self.device_index = device_index
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.tp_group = tp_group
        self.on_gpu = torch.cuda.is_available()
        self.device = torch.device("cuda", device_index) if self.on_gpu \
            else torch.device("cpu")
        if self.on_gpu:
            torch.cuda.set_device(device_index)

        specs = cfg.param_specs(tp_rank, tp_size)
        if slab_bytes is None:
            slab_bytes = actuation.DEFAULT_SLAB_BYTES
        self.actuation_mode = actuation_mode
        t0 = time.perf_counter()
        self.timing = {}
        if actuation_mode == "pack":
            # scattered-tensor mode: parameters live in ordinary caching-
            # allocator storage; sleep/wake goes through the HIP
            # gather/scatter kernel (ops.actuation.PackActuator)
            self.layout, self.total_bytes, _ = actuation.plan_layout(specs)
            self.params = {
                name: torch.empty(shape, dtype=dtype, device=self.device)
                for name, (off, shape, dtype) in self.layout.items()}
            self.arena = None
            self.packer = actuation.PackActuator(
                self.params, mode=pack_xfer_mode, chunk_bytes=chunk_bytes)                 if self.on_gpu else _FakePacker(self.params)
            self.total_bytes = self.packer.total_bytes
        else:
            self.layout, self.total_bytes, slab_sizes = actuation.plan_layout(
                specs, slab_bytes=slab_bytes)
            self.arena = actuation.make_arena(self.total_bytes, device_index,
                                              try_vmm=use_vmm,
                                              chunk_bytes=chunk_bytes,
                                              slab_sizes=slab_sizes,
                                              nstreams=nstreams)
            self.packer = None
            self.params = self._make_views()
        self.timing["arena_s"] = time.perf_counter() - t0
        t1 = time.perf_counter()
        self.model = LlamaModel(cfg, self.params, self.device,
                                tp_rank, tp_size, tp_group)
        if init_weights:
            self.model.init_weights(seed)
        if self.on_gpu:
            torch.cuda.synchronize(self.device)
        self.timing["init_s"] = time.perf_counter() - t1
        t2 = time.perf_counter()
        self.host = actuation.alloc_pinned(self.total_bytes)
        self.timing["pin_s"] = time.perf_counter() - t2
        if self.on_gpu:
            torch.cuda.synchronize(self.device)
        self.create_seconds = time.perf_counter() - t0

        self.state = self.AWAKE
        self._decoder = None  # lazily-captured hipGraph decoder
        self.sleep_count = 0
        self.wake_count = 0
        self.last_sleep_seconds: Optional[float] = None
        self.last_wake_seconds: Optional[float] = None

    # -- lifecycle -----------------------------------------------------------

    def _make_views(self) -> Dict[str, torch.Tensor]:
        return {name: self.arena.view(off, shape, dtype)
                for name, (off, shape, dtype) in self.layout.items()}

    def is_sleeping(self) -> bool:
        return self.state == self.SLEEPING

    def sleep(self, level: int = 1) -> float:
        """Offload weights to pinned host DRAM and free the HBM.

        level=1 == vLLM semantics: weights survive in host memory, KV/cache
        state is discarded (reference README.md:16-26). level=2 would drop
        weights too; we keep the host copy either way (it is the wake
        source) but level is accepted for wire compatibility.
        """
        if self.state == self.SLEEPING:
            return 0.0
        # the graphed decoder pins HBM (its KV cache + graph pool) and its
        # captured pointers die with the arena slabs — drop it first
        self._decoder = None
        if self.packer is not None:
            t = self.packer.sleep(self.host)
        else:
            t = self.arena.sleep(self.host)
        if self.on_gpu:
            # return caching-allocator reserves (activations, KV) so another
            # instance's wake can claim the HBM
            torch.cuda.empty_cache()
        self.state = self.SLEEPING
        self.sleep_count += 1
        self.last_sleep_seconds = t
        return t

    def wake_up(self) -> float:
        """Restore weights to HBM; all TP ranks synchronize before the
        instance reports itself awake."""
        if self.state == self.AWAKE:
            return 0.0
        t0 = time.perf_counter()
        if self.packer is not None:
            self.packer.wake(self.host)
        else:
            _, invalidated = self.arena.wake(self.host)
            if invalidated:
                # non-VMM fallback: arena base moved; re-point the views
                self.params = self._make_views()
                self.model.rebind(self.params)
        if self.tp_size > 1 and dist.is_initialized():
            dist.barrier(group=self.tp_group)
        t = time.perf_counter() - t0
        self.state = self.AWAKE
        self.wake_count += 1
        self.last_wake_seconds = t
        return t

    def load_checkpoint(self, path: str, cpu_threads: int = 8) -> float:
        """Fast checkpoint load: safetensors -> pinned host buffer (parallel
        CPU copies from the mmap) -> one pipelined H2D into the arena.
        Beats tensor-by-tensor pageable copies (~10-20 GB/s) by staging
        through the pinned buffer at the PCIe rate.
        """
        import concurrent.futures
        import time as _t

        if self.actuation_mode != "arena" or self.arena is None:
            from fma_amd.models import loader
            t0 = _t.perf_counter()
            loader.load_into_params(path, self.params,
                                    tp_rank=self.tp_rank,
                                    tp_size=self.tp_size, cfg=self.cfg)
            return _t.perf_counter() - t0

        t0 = _t.perf_counter()
        from fma_amd.models import loader
        host_np = self.host.numpy()

        hf_seen = [False]

        def stage(item):
            name, tensor = item
            if name.startswith("model."):  # HuggingFace checkpoint
                hf_seen[0] = True
                pairs = loader.hf_convert_multi(name, tensor, self.cfg)
                if not pairs:
                    return None  # inv_freq buffers and friends
                if len(pairs) > 1:  # fused tensor split (phi3)
                    return [stage(pr) for pr in pairs]
                name, tensor = pairs[0]
            if name not in self.layout:
                if self.cfg.num_experts and self.cfg.expert_parallel \
                        and ".experts." in name:
                    return None  # another rank's whole expert (EP)
                raise KeyError(f"checkpoint tensor {name!r} unknown")
            off, shape, dtype = self.layout[name]
            tensor = loader.shard_slice(tensor=tensor, name=name,
                                        tp_rank=self.tp_rank,
                                        tp_size=self.tp_size,
                                        local_rows=shape[0] if shape
                                        else None,
                                        local_cols=shape[1]
                                        if len(shape) > 1 else None)
            if tuple(tensor.shape) != tuple(shape):
                raise ValueError(f"shape mismatch for {name}")
            raw = tensor.to(dtype).contiguous().view(torch.uint8).view(-1)
            host_np[off:off + raw.numel()] = raw.numpy()
            return name

        seen = set()
        with concurrent.futures.ThreadPoolExecutor(cpu_threads) as ex:
            for name in ex.map(stage, loader.iter_safetensors(path)):
                if isinstance(name, list):
                    seen.update(n for n in name if n)
                else:
                    seen.add(name)
        missing = set(self.layout) - seen
        if hf_seen[0] and missing == {"lm_head.weight"} \
                and "embed.weight" in seen:
            # transformers tie_word_embeddings: mirror embed's staged
            # bytes into lm_head's slot (both full [vocab, h])
            eo, eshape, edt = self.layout["embed.weight"]
            lo, lshape, ldt = self.layout["lm_head.weight"]
            if eshape == lshape and edt == ldt:
                import math as _m
                nbytes = _m.prod(eshape) * torch.empty(
                    0, dtype=edt).element_size()
                host_np[lo:lo + nbytes] = host_np[eo:eo + nbytes]
                missing = set()
        if missing:
            raise KeyError(f"checkpoint missing parameters: "
                           f"{sorted(missing)[:5]}...")
        self.arena.load_from(self.host)
        if self.on_gpu:
            torch.cuda.synchronize(self.device)
        return _t.perf_counter() - t0

    # -- serving -------------------------------------------------------------

    @torch.no_grad()
    def generate(self, tokens: torch.Tensor, max_new_tokens: int = 16,
                 eos_id: "int | None" = None, temperature: float = 0.0,
                 top_p: float = 1.0) -> torch.Tensor:
        if self.state != self.AWAKE:
            raise RuntimeError("engine is sleeping")
        tokens = tokens.to(self.device)
        if eos_id is None and temperature <= 0.0:
            dec = self._graph_decoder(tokens.shape[0],
                                      tokens.shape[1] + max_new_tokens + 2)
            if dec is not None:
                return dec.generate(tokens, max_new_tokens)
        # EOS stop / sampling are data-dependent: eager decode, not a
        # fixed graph replay
        return self.model.generate(tokens, max_new_tokens, eos_id=eos_id,
                                   temperature=temperature, top_p=top_p)

    def _graph_decoder(self, batch: int, need_seq: int):
        """Lazily built hipGraph decoder for batch-1 serving (2.5x on
        launch-bound models, ~parity on HBM-bound ones; selection via
        FMA_GRAPH_DECODE=1/0/auto). Dropped on sleep: its KV cache holds
        HBM the sleep must release, and the captured kernel args point at
        arena slabs that wake re-allocates — replaying a pre-sleep graph
        after wake would read freed memory. Rebuilt + recaptured at the
        first post-wake generate (~3 decode steps of cost)."""
        mode = os.environ.get("FMA_GRAPH_DECODE", "auto")
        if (mode == "0" or batch != 1 or self.tp_size != 1
                or not self.on_gpu):
            return None
        from fma_amd.models.decode_graph import StaticDecoder as _SD
        if mode != "1" and not _SD.supported(self.cfg):
            # measured ROCm 7.2 graph-replay fault at large vocab
            # (decode_graph.GRAPH_SAFE_VOCAB); eager decode is clean
            return None
        if need_seq > self.cfg.max_seq_len:
            return None
        dec = self._decoder
        if dec is None:
            from fma_amd.models.decode_graph import StaticDecoder
            dec = StaticDecoder(self.model, 1, self.cfg.max_seq_len)
            dec.capture()
            self._decoder = dec
        return dec

    def generate_text(self, prompt: str, max_new_tokens: int = 16,
                      temperature: float = 0.0, top_p: float = 1.0) -> str:
        """Text round trip. With a real tokenizer attached (HF checkpoint
        dirs carry tokenizer.json — see runtime/server.py), prompts
        encode/decode through it; otherwise UTF-8 bytes are the token
        ids (self-contained fallback for synthetic models)."""
        tok = getattr(self, "tokenizer", None)
        if tok is not None:
            ids = tok.encode(prompt).ids or [1]
            toks = torch.tensor([ids], dtype=torch.long, device=self.device)
            out = self.generate(toks, max_new_tokens,
                                eos_id=getattr(self, "eos_id", None),
                                temperature=temperature, top_p=top_p
                                )[0, len(ids):]
            if getattr(self, "eos_id", None) is not None:
                keep = (out == self.eos_id).cumsum(0) == 0
                out = out[keep]
            return tok.decode([int(t) for t in out.tolist()])
        ids = [b % self.cfg.vocab_size for b in prompt.encode("utf-8")] or [1]
        toks = torch.tensor([ids], dtype=torch.long, device=self.device)
        out = self.generate(toks, max_new_tokens)[0, len(ids):]
        return bytes(int(t) % 256 for t in out.tolist()).decode(
            "utf-8", errors="replace")

    def new_kv_cache(self, batch: int, max_seq: Optional[int] = None) -> KVCache:
        return KVCache(self.cfg, batch, self.device, self.tp_size, max_seq)

    def stats(self) -> Dict[str, object]:
        return {
            "model": self.cfg.name,
            "state": self.state,
            "param_bytes": self.total_bytes,
            "tp_rank": self.tp_rank,
            "tp_size": self.tp_size,
            "uses_vmm": getattr(self.arena, "uses_vmm", False),
            "actuation_mode": self.actuation_mode,
            "graph_decode": self._decoder is not None,
            "sleep_count": self.sleep_count,
            "wake_count": self.wake_count,
            "last_sleep_seconds": self.last_sleep_seconds,
            "last_wake_seconds": self.last_wake_seconds,
            "create_seconds": self.create_seconds,
        }


class _FakePacker:
    """CPU emulation of PackActuator for GPU-less tests."""

    def __init__(self, params):
        self.params = params
        self.total_bytes = sum(
            ((p.nbytes + 255) // 256) * 256 for p in params.values())
        self.asleep = False

    def sleep(self, host):
        off = 0
        for name in sorted(self.params):
            p = self.params[name]
            raw = p.contiguous().view(torch.uint8).view(-1)
            host[off:off + raw.numel()].copy_(raw)
            p.zero_()
            off += ((p.nbytes + 255) // 256) * 256
        self.asleep = True
        return 1e-9

    def wake(self, host):
        off = 0
        for name in sorted(self.params):
            p = self.params[name]
            n = p.nbytes
            p.view(torch.uint8).view(-1).copy_(host[off:off + n])
            off += ((n + 255) // 256) * 256
        self.asleep = False
        return 1e-9
