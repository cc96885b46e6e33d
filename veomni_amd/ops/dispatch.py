"""OpSlot — module-level dispatch points used by the modeling code.

API parity target: /root/reference/veomni/ops/dispatch.py:56-131
(`OpSlot(op_name, variant)`, `.bind(impl_name)`, `.use_non_eager_impl`,
`.use_eager_impl`, `.is_bound`, `.bound_kernel()`, `__call__`).
"""

from __future__ import annotations

from typing import Any, Callable, Optional

from .kernel_registry import KERNEL_REGISTRY


class OpSlot:
    def __init__(self, op_name: str, variant: str):
        self.op_name = op_name
        self.variant = variant
        self._kernel: Optional[Callable] = None
        self._impl_name: Optional[str] = None

    def bind(self, impl_name: str) -> None:
        self._kernel = KERNEL_REGISTRY.resolve(self.op_name, self.variant, impl_name)
        self._impl_name = impl_name

    @property
    def use_non_eager_impl(self) -> bool:
        return self._kernel is not None

    @property
    def use_eager_impl(self) -> bool:
        return self._impl_name == "eager"

    @property
    def is_bound(self) -> bool:
        return self._impl_name is not None

    def bound_kernel(self) -> Optional[Callable]:
        return self._kernel

    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        if self._kernel is None:
            raise RuntimeError(
                f"OpSlot('{self.op_name}', '{self.variant}') has no kernel bound."
            )
        return self._kernel(*args, **kwargs)

    def __repr__(self) -> str:
        return f"OpSlot({self.op_name!r}, {self.variant!r}, impl={self._impl_name!r})"
