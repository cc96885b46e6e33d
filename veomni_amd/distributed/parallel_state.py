"""Parallel-state topology for the FSDP2 training step.

API parity target: /root/reference/veomni/distributed/parallel_state.py
(`init_parallel_state` :444-656, `get_parallel_state` :694-701, property set
:60-419). Round-1 scope covers the dims the §8 hot path uses:
(dp_shard, ulysses) as one device mesh, flattened `dp_shard_sp` as the FSDP
shard mesh, and a separate (ep_fsdp, ep) mesh with ep_size | dp_shard_sp
(ref :598-627). pp/tp/cp/dp_replicate are fixed at 1 this round.

MI355X mapping: one process per GPU; the mesh's process groups are RCCL over
xGMI on "cuda" (= ROCm) devices, gloo on CPU for tests.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.distributed as dist
from torch.distributed.device_mesh import DeviceMesh, init_device_mesh

_PARALLEL_STATE: Optional["ParallelState"] = None


@dataclass
class ParallelState:
    world_size: int = 1
    dp_size: int = 1           # dp_shard size
    dp_replicate_size: int = 1  # HSDP outer replication (ref :153-172)
    ulysses_size: int = 1
    ep_size: int = 1
    dp_mode: str = "fsdp2"
    async_ulysses: bool = False
    device_type: str = "cpu"
    device_mesh: Optional[DeviceMesh] = None      # (dp_shard, ulysses)
    ep_device_mesh: Optional[DeviceMesh] = None   # (ep_fsdp, ep)
    _fsdp_mesh: Optional[DeviceMesh] = None       # dp_shard_sp (2-D for HSDP)
    _shard_sp_mesh: Optional[DeviceMesh] = None    # always the flat shard dim
    extra_parallel_names: tuple = ("ep",)

    # ---------------------------------------------------------------- flags
    @property
    def ulysses_enabled(self) -> bool:
        return self.ulysses_size > 1

    @property
    def sp_enabled(self) -> bool:
        # Ulysses is the only SP flavour on the §8 path (cp is dead in the
        # reference too: parallel_state.py:81-82).
        return self.ulysses_enabled

    @property
    def ep_enabled(self) -> bool:
        return self.ep_size > 1

    @property
    def dp_replicate_enabled(self) -> bool:
        return self.dp_replicate_size > 1

    def extra_parallel_enabled(self, name: str) -> bool:
        return name == "ep" and self.ep_enabled

    # ---------------------------------------------------------------- meshes
    @property
    def fsdp_mesh(self) -> Optional[DeviceMesh]:
        return self._fsdp_mesh

    @property
    def fsdp_size(self) -> int:
        # total data-parallel extent FSDP divides gradients by (HSDP
        # includes the replicate dim; ref :242-243 world/(pp*tp))
        return self.dp_replicate_size * self.dp_size * self.ulysses_size

    @property
    def fsdp_shard_size(self) -> int:
        # distinct-shard extent (excludes HSDP replication): the grad-norm
        # p-th-power sum must count each shard once
        return self.dp_size * self.ulysses_size

    @property
    def fsdp_shard_group(self):
        if self._shard_sp_mesh is None:
            return None
        return self._shard_sp_mesh.get_group()

    @property
    def ep_fsdp_mesh(self) -> Optional[DeviceMesh]:
        """Mesh FSDP wraps expert modules over: 2-D (ep_replicate, ep_fsdp)
        under HSDP (HSDP-style fully_shard), 1-D ep_fsdp otherwise."""
        if self.ep_device_mesh is None:
            return None
        if self.dp_replicate_enabled:
            return self.ep_device_mesh["ep_replicate", "ep_fsdp"]
        return self.ep_device_mesh["ep_fsdp"]

    @property
    def ep_shard_group(self):
        """The 1-D ep_fsdp group (distinct expert shards once — the
        grad-norm reduce group; replicas hold identical grads)."""
        if self.ep_device_mesh is None:
            return None
        return self.ep_device_mesh["ep_fsdp"].get_group()

    @property
    def ep_fsdp_size(self) -> int:
        # distinct-shard extent (excludes HSDP replication)
        return self.dp_size * self.ulysses_size // self.ep_size

    # ---------------------------------------------------------------- groups
    @property
    def dp_group(self):
        if self.device_mesh is None:
            return None
        return self.device_mesh["dp_shard"].get_group()

    @property
    def ulysses_group(self):
        if self.device_mesh is None or not self.ulysses_enabled:
            return None
        return self.device_mesh["ulysses"].get_group()

    @property
    def sp_group(self):
        return self.ulysses_group

    @property
    def ep_group(self):
        if self.ep_device_mesh is None:
            return None
        return self.ep_device_mesh["ep"].get_group()

    # ------------------------------------------------------------ ranks/sizes
    @property
    def sp_size(self) -> int:
        return self.ulysses_size

    @property
    def sp_rank(self) -> int:
        return self.ulysses_rank

    @property
    def ulysses_rank(self) -> int:
        if not self.ulysses_enabled:
            return 0
        return dist.get_rank(self.ulysses_group)

    @property
    def ep_rank(self) -> int:
        if not self.ep_enabled:
            return 0
        return dist.get_rank(self.ep_group)

    @property
    def dp_rank(self) -> int:
        if self.device_mesh is None:
            return 0
        return dist.get_rank(self.dp_group)

    # grad divide factor for EP modules — always world size
    # (ref parallel_state.py:348-356).
    def extra_parallel_gradient_divide_factor(self, name: str) -> int:
        return self.world_size

    @property
    def ep_gradient_divide_factor(self) -> int:
        return self.extra_parallel_gradient_divide_factor("ep")


def init_parallel_state(
    dp_size: Optional[int] = None,
    ulysses_size: int = 1,
    ep_size: int = 1,
    dp_mode: str = "fsdp2",
    async_ulysses: bool = False,
    device_type: Optional[str] = None,
    dp_replicate_size: int = 1,
) -> ParallelState:
    """Build the device meshes and register the global state.

    Mirrors the mesh construction of reference parallel_state.py:444-656:
    one mesh (dp_shard, ulysses), a flattened `dp_shard_sp` sub-mesh used as
    the FSDP shard mesh, and a second mesh (ep_fsdp, ep) with the constraint
    ep_size | dp_shard_sp_size (ref :598-627).
    """
    global _PARALLEL_STATE
    if not dist.is_initialized():
        # single-process fast path (no process group, no meshes): the FSDP2
        # wrap short-circuits at fsdp_size == 1 and SP/EP are disabled.
        assert ulysses_size == 1 and ep_size == 1, "multi-dim parallel needs a process group"
        _PARALLEL_STATE = ParallelState(
            world_size=1, dp_size=1, dp_mode=dp_mode,
            device_type=device_type or ("cuda" if torch.cuda.is_available() else "cpu"),
        )
        return _PARALLEL_STATE
    world_size = dist.get_world_size()
    if device_type is None:
        device_type = "cuda" if torch.cuda.is_available() else "cpu"
    if dp_size is None:
        assert world_size % (ulysses_size * dp_replicate_size) == 0
        dp_size = world_size // (ulysses_size * dp_replicate_size)
    assert dp_replicate_size * dp_size * ulysses_size == world_size, (
        dp_replicate_size, dp_size, ulysses_size, world_size)
    dp_shard_sp = dp_size * ulysses_size
    assert dp_shard_sp % ep_size == 0, f"ep_size {ep_size} must divide dp_shard*sp {dp_shard_sp}"

    if dp_replicate_size > 1:
        # HSDP: outer replicate dim; FSDP consumes the 2-D
        # (dp_replicate, dp_shard_sp) mesh (ref fsdp_mesh property :218-231)
        mesh = init_device_mesh(
            device_type, (dp_replicate_size, dp_size, ulysses_size),
            mesh_dim_names=("dp_replicate", "dp_shard", "ulysses"))
        shard_sp_mesh = mesh["dp_shard", "ulysses"]._flatten(mesh_dim_name="dp_shard_sp")
        fsdp_mesh = mesh["dp_replicate", "dp_shard_sp"]
    else:
        mesh = init_device_mesh(
            device_type, (dp_size, ulysses_size), mesh_dim_names=("dp_shard", "ulysses")
        )
        fsdp_mesh = mesh["dp_shard", "ulysses"]._flatten(mesh_dim_name="dp_shard_sp")
        shard_sp_mesh = fsdp_mesh

    ep_mesh = None
    if ep_size > 1:
        if dp_replicate_size > 1:
            # HSDP + EP (ref :605-627): experts replicate with the dense
            # outer dim -> mesh (ep_replicate, ep_fsdp, ep)
            ep_mesh = init_device_mesh(
                device_type, (dp_replicate_size, dp_shard_sp // ep_size, ep_size),
                mesh_dim_names=("ep_replicate", "ep_fsdp", "ep"))
        else:
            ep_mesh = init_device_mesh(
                device_type, (dp_shard_sp // ep_size, ep_size), mesh_dim_names=("ep_fsdp", "ep")
            )

    _PARALLEL_STATE = ParallelState(
        world_size=world_size,
        dp_size=dp_size,
        dp_replicate_size=dp_replicate_size,
        ulysses_size=ulysses_size,
        ep_size=ep_size,
        dp_mode=dp_mode,
        async_ulysses=async_ulysses,
        device_type=device_type,
        device_mesh=mesh,
        ep_device_mesh=ep_mesh,
        _fsdp_mesh=fsdp_mesh,
        _shard_sp_mesh=shard_sp_mesh,
    )
    return _PARALLEL_STATE


def get_parallel_state() -> ParallelState:
    """Current state; an uninitialized process gets a single-process default
    (ref parallel_state.py:694-701)."""
    global _PARALLEL_STATE
    if _PARALLEL_STATE is None:
        return ParallelState()
    return _PARALLEL_STATE


def set_parallel_state(state: Optional[ParallelState]) -> None:
    global _PARALLEL_STATE
    _PARALLEL_STATE = state
