"""Operator API (drop-in mirror of the reference's `veomni.ops`).

Importing this package registers every MI355X kernel under impl name "hip"
in KERNEL_REGISTRY. `HIP_OPS_CONFIG` is the per-op implementation map the
bench/tests bind (the OpsImplementationConfig analog)."""

from .dispatch import OpSlot  # noqa: F401
from .kernel_registry import (  # noqa: F401
    KERNEL_REGISTRY,
    HardwareRequirement,
    KernelSpec,
)
from . import kernels  # noqa: F401  (registrations run at import)

# Per-op implementation selection for the MI355X hot path. The aux
# load-balancing loss stays host-side torch this round: it is a tiny [T, E]
# reduction off the §8 critical path (SURVEY §8a "small").
HIP_OPS_CONFIG = {
    "rms_norm": "hip",
    "rotary_pos_emb": "hip",
    "swiglu_mlp": "hip",
    "moe_experts": "hip",
    "cross_entropy_loss": "hip",
    "load_balancing_loss": "eager",
    "attention": "hip",
}
