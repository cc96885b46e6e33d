"""veomni_amd — MI355X-native rebuild of VeOmni's FSDP2 training hot path.

Drop-in for the reference's `veomni.ops` operator API and `veomni.distributed`
parallel API. All device compute is hand-written gfx950 HIP
behind the C ABI in include/veomni_hip.h (`libveomni_hip.so`); the host side
is Python on PyTorch-ROCm, with torch.distributed (RCCL over xGMI) as the
collective transport.
"""

__version__ = "0.1.0"
