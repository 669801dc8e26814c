"""fma_amd — MI355X-native fast model actuation.

A from-scratch stack with the capabilities of
llm-d-incubation/llm-d-fast-model-actuation, re-designed for AMD MI355X:
the sleep/wake hot path is a HIP (CDNA4/gfx950) pack/scatter + pinned-DRAM
transfer library, multi-GPU wake is coordinated over RCCL/xGMI, and the
control plane (dual-pods controllers, launcher, requester) speaks the same
CRDs, annotations and HTTP contracts as the reference.
"""

__version__ = "0.2.0"
