"""Flagship benchmark: wake_up latency for a 64 GiB model on MI355X.

Measures BASELINE.json's headline metric — "wake_up latency (s) +
time-to-ready after swap, 64 GiB model @ 1/2/4/8 MI355X" — against the
reference's published ≈3 s for 64 GiB of tensors (reference README.md:24-25,
other hardware).

One step = one full sleep(level=1) -> wake_up() actuation cycle of a
70B-shaped synthetic Llama with 64 GiB of bf16 parameters (random-init;
no network for checkpoints). With N GPUs the model is TP-sharded (strong
scaling: fixed 64 GiB total, per-rank bytes ~1/N), each rank packs/restores
its own shard through pinned host DRAM, and an RCCL barrier over xGMI gates
the wake so all ranks re-enter serving together. The reported value is the
mean wake_up latency in seconds (max over ranks) — time-to-ready, since
the engine flips is_sleeping only after the barrier.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N ... bench.py --gpus N ...
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def log(msg: str) -> None:
    print(msg, file=sys.stderr, flush=True)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--gib", type=float, default=64.0,
                    help="total parameter GiB across all ranks")
    # pack (the gather/scatter-kernel path BASELINE.json's north star
    # describes) reached arena parity in round 2 (1.2175 vs 1.2165 s /
    # 64 GiB) and is now the measured default
    ap.add_argument("--mode", choices=["arena", "pack"], default="pack")
    ap.add_argument("--pack-xfer", type=int, default=-1,
                    help="pack transfer mode: -1 auto (default; measured "
                         "fastest per direction), 0 staged-kernel, "
                         "1 direct-kernel, 2 per-tensor")
    ap.add_argument("--chunk-mb", type=int, default=0)
    ap.add_argument("--nstreams", type=int, default=1)
    ap.add_argument("--slab-mb", type=int, default=0, help="slab size MiB (0=default 1 GiB)")
    ap.add_argument("--vmm", action="store_true", help="opt into VMM arena (unreliable on ROCm 7.2)")
    ap.add_argument("--no-swap", action="store_true",
                    help="skip the model-swap time-to-ready measurement")
    args = ap.parse_args()

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from fma_amd.models.llama import LlamaConfig
    from fma_amd.runtime.engine import ActuationEngine

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    on_gpu = torch.cuda.is_available()
    dist_on = world > 1
    if dist_on:
        from datetime import timedelta

        import torch.distributed as dist
        backend = os.environ.get("FMA_DIST_BACKEND") or (
            "nccl" if on_gpu else "gloo")
        # RCCL preflight: dmabuf IPC is the only mode the pool's driver
        # supports; a hung rank should fail the job in minutes, not hang
        # the box until the driver's limit kills it
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        os.environ.setdefault("NCCL_DEBUG", "WARN")
        os.environ.setdefault("TORCH_NCCL_HEARTBEAT_TIMEOUT_SEC", "180")
        if on_gpu:
            ngpu = torch.cuda.device_count()
            if local_rank >= ngpu:
                if backend == "gloo":
                    local_rank = local_rank % ngpu  # rehearsal on 1 GPU
                else:
                    log(f"[rank {rank}] FATAL: LOCAL_RANK {local_rank} "
                        f"but only {ngpu} visible GPUs")
                    sys.exit(2)
            torch.cuda.set_device(local_rank)
        try:
            dist.init_process_group(backend=backend, rank=rank,
                                    world_size=world,
                                    timeout=timedelta(seconds=180))
        except Exception as e:  # noqa: BLE001 - surface, don't hang
            log(f"[rank {rank}] FATAL: init_process_group({backend}) "
                f"failed: {e}")
            sys.exit(2)
    n_gpus = world if dist_on else args.gpus
    assert n_gpus == world or world == 1, \
        f"--gpus {args.gpus} vs WORLD_SIZE {world} mismatch"

    cfg = LlamaConfig.from_total_gib(args.gib)
    t0 = time.perf_counter()
    eng = ActuationEngine(
        cfg, device_index=local_rank if on_gpu else 0,
        tp_rank=rank, tp_size=world, tp_group=None,
        use_vmm=args.vmm or None, chunk_bytes=args.chunk_mb << 20, seed=1234,
        nstreams=args.nstreams,
        slab_bytes=(args.slab_mb << 20) if args.slab_mb else None,
        actuation_mode=args.mode,
        pack_xfer_mode=None if args.pack_xfer < 0 else args.pack_xfer)
    log(f"[rank {rank}] engine up: {eng.total_bytes/2**30:.2f} GiB/rank, "
        f"{cfg.num_layers} layers, vmm={eng.stats()['uses_vmm']}, "
        f"create {time.perf_counter()-t0:.1f}s")

    def barrier():
        if dist_on:
            import torch.distributed as dist
            dist.barrier()

    def sync():
        if on_gpu:
            torch.cuda.synchronize()

    def alloc_wait():
        return getattr(eng.arena, "last_alloc_wait_seconds", 0.0) \
            if eng.arena is not None else 0.0

    # Box health probe: raw pinned H2D bandwidth + hipMalloc commit rate.
    # Wake is link-bound (~56 GB/s on a healthy PCIe Gen5 x16 host); some
    # pool hosts have degraded H2D or slow page commit (seen: 20 GB/s ->
    # 3.1 s instead of 1.22 s for 64 GiB). Reporting both makes a slow
    # box self-evident in the result rather than looking like a
    # regression.
    h2d_gbps = alloc_gbps = None
    if on_gpu:
        probe = min(1 << 30, eng.total_bytes or (1 << 30))
        host_buf = torch.empty(probe, dtype=torch.uint8, pin_memory=True)
        dev_buf = torch.empty(probe, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        t0p = time.perf_counter()
        dev_buf.copy_(host_buf, non_blocking=True)
        torch.cuda.synchronize()
        h2d_gbps = probe / (time.perf_counter() - t0p) / 1e9
        del dev_buf
        torch.cuda.empty_cache()
        t0p = time.perf_counter()
        dev_buf = torch.empty(probe, dtype=torch.uint8, device="cuda")
        dev_buf.fill_(1)  # force page commit
        torch.cuda.synchronize()
        alloc_gbps = probe / (time.perf_counter() - t0p) / 1e9
        del dev_buf, host_buf
        torch.cuda.empty_cache()
        log(f"[rank {rank}] box probe: pinned H2D {h2d_gbps:.1f} GB/s, "
            f"alloc+commit {alloc_gbps:.1f} GB/s"
            + ("  << DEGRADED HOST (healthy: ~56 GB/s H2D)"
               if h2d_gbps < 40 else ""))
        if dist_on:
            # report the WORST rank's link: wake is gated by the slowest
            import torch.distributed as dist
            dev = torch.device("cuda", local_rank)
            worst = torch.tensor([h2d_gbps, alloc_gbps],
                                 dtype=torch.float64, device=dev)
            dist.all_reduce(worst, op=dist.ReduceOp.MIN)
            h2d_gbps, alloc_gbps = worst.tolist()

    # warmup
    for i in range(args.warmup):
        ts = eng.sleep()
        tw = eng.wake_up()
        log(f"[rank {rank}] warmup {i}: sleep {ts:.3f}s wake {tw:.3f}s "
            f"(alloc-wait {alloc_wait():.3f}s)")

    barrier()
    sync()
    wall0 = time.perf_counter()
    wake_times = []
    sleep_times = []
    for _ in range(args.steps):
        sleep_times.append(eng.sleep())
        wake_times.append(eng.wake_up())
        log(f"[rank {rank}] step: sleep {sleep_times[-1]:.3f}s "
            f"wake {wake_times[-1]:.3f}s (alloc-wait {alloc_wait():.3f}s)")
    barrier()
    sync()
    wall1 = time.perf_counter()

    # time-to-ready after swap (the second half of the metric string):
    # a second same-shaped model is parked asleep in pinned host DRAM;
    # one swap = sleep(A) -> wake(B) -> B ready (the launcher model-swap,
    # BASELINE config #3, measured at flagship size). Engine B is built
    # after the timed region so its random-init cost stays out of both.
    swap_ready_s = None
    if not args.no_swap:
        eng_b = ActuationEngine(
            cfg, device_index=local_rank if on_gpu else 0,
            tp_rank=rank, tp_size=world, tp_group=None,
            use_vmm=args.vmm or None, chunk_bytes=args.chunk_mb << 20,
            seed=4321, nstreams=args.nstreams,
            slab_bytes=(args.slab_mb << 20) if args.slab_mb else None,
            actuation_mode=args.mode,
            pack_xfer_mode=None if args.pack_xfer < 0 else args.pack_xfer)
        eng_b.sleep()  # park B in host DRAM
        barrier()
        sync()
        t0s = time.perf_counter()
        eng.sleep()
        eng_b.wake_up()
        barrier()
        sync()
        swap_ready_s = time.perf_counter() - t0s
        log(f"[rank {rank}] swap A->B ready in {swap_ready_s:.3f}s")
        eng_b.sleep()
        eng.wake_up()  # restore A so repeated runs see identical state

    ms_per_step = (wall1 - wall0) / args.steps * 1000.0
    mean_wake = sum(wake_times) / len(wake_times)
    mean_sleep = sum(sleep_times) / len(sleep_times)

    if dist_on:
        import torch.distributed as dist
        dev = torch.device("cuda", local_rank) if on_gpu else torch.device("cpu")
        agg = torch.tensor([ms_per_step, mean_wake, mean_sleep,
                            swap_ready_s or 0.0],
                           dtype=torch.float64, device=dev)
        dist.all_reduce(agg, op=dist.ReduceOp.MAX)
        ms_per_step, mean_wake, mean_sleep, swap_max = agg.tolist()
        if swap_ready_s is not None:
            swap_ready_s = swap_max

    if rank == 0:
        baseline_s = 3.0  # reference: ~3 s wake for 64 GiB (README.md:24-25)
        value = mean_wake
        result = {
            "metric": "wake_up latency (s) + time-to-ready after swap, "
                      "64 GiB model @ 1/2/4/8 MI355X",
            "value": round(value, 4),
            "unit": "s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": round(value / baseline_s, 4) if args.gib == 64 else None,
            "dtype": "bf16",
            "data": "synthetic (random-init weights, 70B-shaped llama)",
            "config": {
                "model": cfg.name,
                "param_gib_total": args.gib,
                "param_gib_per_rank": round(eng.total_bytes / 2**30, 3),
                "layers": cfg.num_layers,
                "parallelism": f"tp{n_gpus}",
                "mode": args.mode,
                "vmm": bool(eng.stats()["uses_vmm"]),
                "mean_sleep_s": round(mean_sleep, 4),
                "swap_ready_s": round(swap_ready_s, 4)
                if swap_ready_s is not None else None,
                "h2d_gbps": round(h2d_gbps, 1) if h2d_gbps else None,
                "alloc_gbps": round(alloc_gbps, 1) if alloc_gbps else None,
                "global_batch": None,
                "seq_len": None,
            },
        }
        print(json.dumps(result), flush=True)

    if dist_on:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
