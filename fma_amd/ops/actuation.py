"""Python API over the MI355X actuator extension (fma_amd._C).

Two actuation strategies, both moving model tensors between GPU HBM3E and
pinned host DRAM (the native replacement for vLLM's sleep level=1 the
reference drives over HTTP — reference README.md:16-26,
pkg/controller/dual-pods/inference-server.go:1497,1712):

- :class:`ArenaActuator` — parameters live as views into one contiguous
  device arena, so sleep/wake is a pure chunked pinned-memcpy (plus, in VMM
  mode, physical map/unmap at a constant virtual address). This is the fast
  path used by the serving runtime for models it materializes itself.
- :class:`PackActuator` — for tensors scattered across the caching
  allocator: the HIP gather kernel coalesces every shard into one flat
  space for a single pinned D2H per chunk; wake re-allocates the storages
  and scatters back.

On machines without a GPU (CI), :class:`FakeArenaActuator` emulates the
same interface on CPU memory so the serving runtime, launcher and
controllers are fully testable (the mock seam the reference achieves with
its CPU vLLM image — reference docs/launcher.md:656-707).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Sequence, Tuple

import torch

ARENA_ALIGN = 256

# staged / direct / per_tensor map to XferMode in csrc/actuator.cpp
MODE_STAGED = 0
MODE_DIRECT = 1
MODE_PER_TENSOR = 2

_C = None
_IMPORT_ERROR: Optional[BaseException] = None
try:
    from fma_amd import _C as _C  # type: ignore[attr-defined, no-redef]
except Exception as e:  # pragma: no cover - exercised only when ext missing
    _IMPORT_ERROR = e


def native_available() -> bool:
    return _C is not None


def require_native():
    """The HIP extension is mandatory whenever a GPU is present: silently
    falling back to an eager path on real hardware would invalidate every
    latency number, so fail loudly instead."""
    if _C is None:
        raise ImportError(
            "fma_amd._C (HIP actuator extension) is not built; run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_IMPORT_ERROR!r}"
        )
    return _C


def align_up(n: int, a: int = ARENA_ALIGN) -> int:
    return (n + a - 1) // a * a


#: default slab size for the segmented arena: big enough that per-slab
#: hipMalloc overhead amortizes, small enough that wake_up overlaps most
#: allocation time behind in-flight PCIe copies.
DEFAULT_SLAB_BYTES = 512 << 20


def plan_layout(specs: Sequence[Tuple[str, Tuple[int, ...], torch.dtype]],
                slab_bytes: int = 0,
                ) -> Tuple[Dict[str, Tuple[int, Tuple[int, ...], torch.dtype]],
                           int, List[int]]:
    """Assign 256-B-aligned flat offsets to named tensor specs, packing them
    into slabs of ~slab_bytes so no tensor straddles a slab boundary
    (slab_bytes=0: one slab).

    Returns ({name: (offset, shape, dtype)}, total_bytes, slab_sizes).
    """
    layout: Dict[str, Tuple[int, Tuple[int, ...], torch.dtype]] = {}
    slab_sizes: List[int] = []
    closed = 0   # total bytes in closed slabs
    used = 0     # bytes used in the open slab
    for name, shape, dtype in specs:
        numel = 1
        for s in shape:
            numel *= s
        nbytes = align_up(max(numel * torch.empty(0, dtype=dtype).element_size(), 1))
        if slab_bytes and used and used + nbytes > slab_bytes:
            slab_sizes.append(used)
            closed += used
            used = 0
        layout[name] = (closed + used, tuple(shape), dtype)
        used += nbytes
    if used:
        slab_sizes.append(used)
        closed += used
    return layout, closed, slab_sizes


def alloc_pinned(nbytes: int) -> torch.Tensor:
    """One pinned host buffer. Allocated once per instance and reused across
    every sleep/wake cycle (pinning cost stays off the wake path). Plain
    host memory on GPU-less machines (tests)."""
    if torch.cuda.is_available():
        return torch.empty(nbytes, dtype=torch.uint8, pin_memory=True)
    return torch.empty(nbytes, dtype=torch.uint8)


class ArenaActuator:
    """Contiguous device arena with sleep/wake via one pinned host buffer."""

    def __init__(self, nbytes: int, device: int = 0,
                 try_vmm: Optional[bool] = None, chunk_bytes: int = 0,
                 slab_sizes: Optional[List[int]] = None, nstreams: int = 1):
        C = require_native()
        if try_vmm is None:
            # VMM (constant-VA remap) measured UNRELIABLE on ROCm 7.2 /
            # gfx950: after unmap→remap or VA reuse, SDMA reads are
            # intermittently stale (tools/debug_arena.py: 86-100% corrupt
            # cycles under swap stress). Default is the slabbed hipMalloc
            # arena — views are re-bound on wake, which costs milliseconds.
            try_vmm = os.environ.get("FMA_TRY_VMM") == "1"
        slabs = list(slab_sizes or [])
        if try_vmm:
            try:
                self._arena = C.DeviceArena(nbytes, device, True, [])
            except RuntimeError as e:
                import warnings
                warnings.warn(f"VMM arena failed ({e}); falling back to "
                              "slabbed hipMalloc arena (views re-bound on wake)")
                self._arena = C.DeviceArena(nbytes, device, False, slabs)
        else:
            self._arena = C.DeviceArena(nbytes, device, False, slabs)
        self.device = device
        self.nbytes = nbytes
        self.chunk_bytes = chunk_bytes
        self.nstreams = nstreams
        self._gen_at_view: int = self._arena.generation
        threads = int(os.environ.get("FMA_ALLOC_THREADS", "4"))
        self._arena.set_alloc_threads(threads)

    @property
    def uses_vmm(self) -> bool:
        return self._arena.uses_vmm

    @property
    def is_mapped(self) -> bool:
        return self._arena.is_mapped

    @property
    def last_map_seconds(self) -> float:
        return self._arena.last_map_seconds

    @property
    def last_alloc_wait_seconds(self) -> float:
        return self._arena.last_alloc_wait_seconds

    def view(self, offset: int, shape: Tuple[int, ...], dtype: torch.dtype
             ) -> torch.Tensor:
        return self._arena.view(offset, list(shape), dtype)

    def sleep(self, host: torch.Tensor) -> float:
        return self._arena.sleep_to(host, self.chunk_bytes, self.nstreams)

    def load_from(self, host: torch.Tensor) -> float:
        """Refill a mapped arena from the pinned buffer (checkpoint load)."""
        return self._arena.load_from(host, self.chunk_bytes, self.nstreams)

    def wake(self, host: torch.Tensor) -> Tuple[float, bool]:
        """Returns (seconds, views_invalidated). With VMM backing the VA is
        constant and views survive; otherwise the caller must re-bind."""
        t = self._arena.wake_from(host, self.chunk_bytes, self.nstreams)
        self._gen_at_view = self._arena.generation
        # VMM keeps the VA constant; slabbed/plain arenas re-allocate, so
        # every wake invalidates existing views.
        return t, not self._arena.uses_vmm


class FakeArenaActuator:
    """CPU emulation of ArenaActuator for GPU-less tests.

    Guarded by FMA_FAKE_GPU=1 (or explicit construction) so a missing
    extension on a real GPU box can never silently fall through to this.
    """

    def __init__(self, nbytes: int, device: int = 0, try_vmm=None,
                 chunk_bytes: int = 0, slab_sizes=None, nstreams: int = 1):
        if not (os.environ.get("FMA_FAKE_GPU") == "1"
                or not torch.cuda.is_available()):
            raise RuntimeError("FakeArenaActuator is only for GPU-less machines")
        self.device = device
        self.nbytes = nbytes
        self.chunk_bytes = chunk_bytes
        self._buf = torch.zeros(nbytes, dtype=torch.uint8)
        self._asleep = False
        self.uses_vmm = True  # emulates the constant-VA behavior

    @property
    def is_mapped(self) -> bool:
        return not self._asleep

    def view(self, offset: int, shape: Tuple[int, ...], dtype: torch.dtype
             ) -> torch.Tensor:
        assert not self._asleep, "arena is asleep"
        numel = 1
        for s in shape:
            numel *= s
        nbytes = numel * torch.empty(0, dtype=dtype).element_size()
        return self._buf[offset:offset + nbytes].view(dtype).view(shape)

    def sleep(self, host: torch.Tensor) -> float:
        assert not self._asleep, "arena already asleep"
        host[: self.nbytes].copy_(self._buf)
        self._buf.zero_()  # poison: reads while asleep are wrong by design
        self._asleep = True
        return 1e-9

    def load_from(self, host: torch.Tensor) -> float:
        assert not self._asleep, "arena must be awake to load into"
        self._buf.copy_(host[: self.nbytes])
        return 1e-9

    def wake(self, host: torch.Tensor) -> Tuple[float, bool]:
        assert self._asleep, "arena already awake"
        self._buf.copy_(host[: self.nbytes])
        self._asleep = False
        return 1e-9, False


class PackActuator:
    """Sleep/wake for tensors scattered across the caching allocator.

    sleep: HIP gather kernel -> staging -> pinned host, then storages are
    resized to zero and the caching allocator's reserve is returned to the
    system so another instance can use the HBM.
    wake: storages re-allocated, pinned host -> staging -> HIP scatter.
    """

    def __init__(self, tensors: Dict[str, torch.Tensor],
                 mode: Optional[int] = None, chunk_bytes: int = 0):
        require_native()
        self.names: List[str] = sorted(tensors.keys())
        self.tensors = tensors
        mean = (sum(t.nbytes for t in tensors.values())
                / max(len(tensors), 1))
        if mode is None:
            # measured on MI355X (tools/cycle_probe.py): the gather kernel
            # pays off for many small tensors (per-copy SDMA overhead would
            # dominate); for large shards per-tensor SDMA wins — and for
            # SLEEP specifically, re-allocated caching-allocator memory
            # reads back through the kernel at ~34 GiB/s (TLB-unfriendly
            # page granularity after empty_cache) while SDMA reads stay at
            # ~52 GiB/s, so sleep prefers SDMA once shards are big enough
            mode = MODE_PER_TENSOR if mean >= (16 << 20) else MODE_STAGED
            self.sleep_mode = mode
        else:
            self.sleep_mode = mode
        self.mode = mode
        self.chunk_bytes = chunk_bytes
        off = 0
        self.offsets: List[int] = []
        for n in self.names:
            self.offsets.append(off)
            off += align_up(max(tensors[n].nbytes, 1))
        self.total_bytes = off
        self.asleep = False

    def _tensor_list(self) -> List[torch.Tensor]:
        return [self.tensors[n] for n in self.names]

    def sleep(self, host: torch.Tensor) -> float:
        C = require_native()
        t = C.pack_to_host(self._tensor_list(), self.offsets, host,
                           self.sleep_mode, self.chunk_bytes)
        for n in self.names:
            self.tensors[n].untyped_storage().resize_(0)
        torch.cuda.empty_cache()
        self.asleep = True
        return t

    def wake(self, host: torch.Tensor) -> float:
        C = require_native()
        if self.mode in (MODE_STAGED, MODE_PER_TENSOR) and \
                hasattr(C, "restore_from_host_overlapped"):
            # storage re-allocation happens on a background thread inside
            # the call, overlapped with the H2D pipeline (the up-front
            # resize loop cost ~0.3 s serial on a 64 GiB model)
            t = C.restore_from_host_overlapped(
                self._tensor_list(), self.offsets, host, self.mode,
                self.chunk_bytes)
        else:
            for n in self.names:
                tt = self.tensors[n]
                tt.untyped_storage().resize_(tt.numel() * tt.element_size())
            t = C.restore_from_host(self._tensor_list(), self.offsets, host,
                                    self.mode, self.chunk_bytes)
        self.asleep = False
        return t


def make_arena(nbytes: int, device: int = 0, try_vmm=None,
               chunk_bytes: int = 0, slab_sizes=None, nstreams: int = 1):
    """Arena factory: native on a GPU machine, fake on CPU-only machines."""
    if torch.cuda.is_available():
        return ArenaActuator(nbytes, device, try_vmm, chunk_bytes,
                             slab_sizes, nstreams)
    return FakeArenaActuator(nbytes, device, try_vmm, chunk_bytes,
                             slab_sizes, nstreams)
