from . import attention, cross_entropy, moe, norms  # noqa: F401
