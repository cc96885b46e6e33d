from .fsdp2 import build_parallelize_model, clip_grad_norm  # noqa: F401
from .parallel_plan import ParallelPlan  # noqa: F401
from .parallel_state import (  # noqa: F401
    ParallelState,
    get_parallel_state,
    init_parallel_state,
    set_parallel_state,
)
from .sequence_parallel import (  # noqa: F401
    gather_heads_scatter_seq,
    gather_seq_scatter_heads,
    reduce_sequence_parallel_loss,
    set_ulysses_sequence_parallel_group,
)
