"""Stress arena sleep/wake bit-exactness on a GPU box (flakiness hunter)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import fma_amd._C as C  # noqa: E402


def cycle_check(arena, host, n, chunk, tag):
    v = arena.view(0, [n], torch.bfloat16)
    v.normal_()
    torch.cuda.synchronize()
    snap = v.clone()
    torch.cuda.synchronize()
    arena.sleep_to(host, chunk)
    href = host[:n * 2].view(torch.bfloat16)
    host_mm = int((href != snap.cpu()).sum())
    arena.wake_from(host, chunk)
    v2 = arena.view(0, [n], torch.bfloat16)
    dev_mm = int((v2 != snap).sum())
    old_mm = int((v != snap).sum())
    if host_mm or dev_mm or old_mm:
        print(f"  [{tag}] MISMATCH host={host_mm} fresh_view={dev_mm} "
              f"old_view={old_mm} of {n}")
        return False
    return True


def run(nbytes, chunk, vmm, reps=10):
    arena = C.DeviceArena(nbytes, 0, vmm)
    host = torch.empty(nbytes, dtype=torch.uint8, pin_memory=True)
    n = nbytes // 2
    ok = sum(cycle_check(arena, host, n, chunk,
                         f"vmm={vmm} chunk={chunk>>20}M rep={r}")
             for r in range(reps))
    print(f"nbytes={nbytes>>20}MiB chunk={chunk>>20}MiB vmm={vmm}: "
          f"{ok}/{reps} clean")
    del arena


def swap_stress(reps=6, vmm=True):
    """Two arenas alternating sleep/wake (model-swap pattern, VA churn)."""
    nb = 128 << 20
    a1 = C.DeviceArena(nb, 0, vmm)
    h1 = torch.empty(nb, dtype=torch.uint8, pin_memory=True)
    a2 = C.DeviceArena(nb, 0, vmm)
    h2 = torch.empty(nb, dtype=torch.uint8, pin_memory=True)
    v1 = a1.view(0, [nb // 2], torch.bfloat16)
    v2 = a2.view(0, [nb // 2], torch.bfloat16)
    v1.normal_(); v2.normal_()
    torch.cuda.synchronize()
    s1, s2 = v1.clone(), v2.clone()
    torch.cuda.synchronize()
    a2.sleep_to(h2, 0)
    clean = 0
    for r in range(reps):
        a1.sleep_to(h1, 0)
        a2.wake_from(h2, 0)
        mm2 = int((a2.view(0, [nb // 2], torch.bfloat16) != s2).sum())
        a2.sleep_to(h2, 0)
        a1.wake_from(h1, 0)
        mm1 = int((a1.view(0, [nb // 2], torch.bfloat16) != s1).sum())
        if mm1 or mm2:
            print(f"  swap rep {r}: mm1={mm1} mm2={mm2}")
        else:
            clean += 1
    print(f"swap_stress vmm={vmm}: {clean}/{reps} clean")


if __name__ == "__main__":
    for vmm in (True, False):
        for chunk in (0, 8 << 20):
            run(64 << 20, chunk, vmm, reps=10)
    run(1 << 30, 0, True, reps=5)
    swap_stress(vmm=True)
    swap_stress(vmm=False)
    print("done")
