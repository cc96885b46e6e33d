"""Kernel registry — mirror of the reference operator-provider API.

API parity target: /root/reference/veomni/ops/kernel_registry.py:34-172
(`HardwareRequirement`, `KernelSpec(name, op_name, variant, factory,
hardware, description)`, `KERNEL_REGISTRY.register/resolve/list_available`,
`resolve(..., "eager") -> None`). The MI355X build registers its kernels
under impl name "hip"; "eager" always means the model's own host code.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Optional

import torch


@dataclass(frozen=True)
class HardwareRequirement:
    device_type: str  # "gpu" (a ROCm device) | "any" (pure torch)

    def is_satisfied(self) -> bool:
        if self.device_type == "gpu":
            return torch.cuda.is_available()
        if self.device_type == "any":
            return True
        raise ValueError(f"Unknown device_type: {self.device_type!r} (expected 'gpu' | 'any')")


@dataclass(frozen=True)
class KernelSpec:
    name: str
    op_name: str
    variant: str
    factory: Callable[[], Callable]
    hardware: HardwareRequirement
    description: str = ""


class KernelRegistry:
    """(op_name, variant) -> {impl_name: KernelSpec}."""

    def __init__(self):
        self._specs: dict[tuple[str, str], dict[str, KernelSpec]] = {}

    def register(self, spec: KernelSpec, force: bool = False) -> None:
        bucket = self._specs.setdefault((spec.op_name, spec.variant), {})
        if spec.name in bucket and not force:
            raise ValueError(
                f"Duplicate kernel registration: op='{spec.op_name}', "
                f"variant='{spec.variant}', name='{spec.name}'"
            )
        bucket[spec.name] = spec

    def resolve(self, op_name: str, variant: str, impl_name: str) -> Optional[Callable]:
        if impl_name == "eager":
            return None
        bucket = self._specs.get((op_name, variant), {})
        if impl_name not in bucket:
            available = list(bucket.keys()) + ["eager"]
            raise KeyError(
                f"Unknown kernel '{impl_name}' for op='{op_name}', variant='{variant}'. "
                f"Available: {available}"
            )
        spec = bucket[impl_name]
        if not spec.hardware.is_satisfied():
            raise RuntimeError(
                f"Kernel '{impl_name}' for op='{op_name}' requires "
                f"device_type='{spec.hardware.device_type}', but the current "
                "hardware does not satisfy this."
            )
        return spec.factory()

    def list_available(self, op_name: str, variant: str) -> list[str]:
        return list(self._specs.get((op_name, variant), {}).keys())


KERNEL_REGISTRY = KernelRegistry()
