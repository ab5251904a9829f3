# ORACLE — TEST INFRASTRUCTURE ONLY.
#
# This package is a CPU restatement of the reference's attention algorithm
# (SandAI-org/MagiAttention) used exclusively as the parity checker for the
# MI355X HIP kernels. It must ONLY be imported by:
#   - tests/               (parity tests)
#   - __graft_entry__.smoke()   (result check)
#   - bench.py             (the `cpu_baseline` leg)
# The product package `magi_attention/` must never import or route through
# this code; the HIP extension failing to load is a hard error there.
from .ref_attn import make_attn_mask, ref_attn, ref_attn_with_grads  # noqa: F401
