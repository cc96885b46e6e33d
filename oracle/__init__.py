"""CPU oracle for the VeOmni FSDP2-step hot path — TEST INFRASTRUCTURE ONLY.

This package is a CPU restatement of the reference's algorithms
(ByteDance-Seed/VeOmni, mounted read-only at /root/reference during the
build), used exclusively as the parity checker. Only `tests/`,
`__graft_entry__.smoke()` and `bench.py`'s `cpu_baseline` leg may import or
call anything in here. The product path (`veomni_amd/`) must never route
through this package; "hip"-registered kernels fail loudly when the HIP
extension is missing.

Pinning: every function cites the reference file:line it restates, and
`tests/golden/make_golden.py` checks the oracle against the reference's own
eager code executed in the build container (the reference is importable
Python; goldens + the generating script are committed under tests/golden/).
Integer/index work is bit-exact; floating-point work is fp32 with the
tolerances stated in the tests.
"""

from . import losses, moe, norms  # noqa: F401
