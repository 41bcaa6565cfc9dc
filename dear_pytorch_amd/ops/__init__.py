"""Native HIP/CDNA4 ops + communicator bindings.

`comm_core` (RCCL communicator) and `_kernels` (fused update / pack / topk
kernels) are in-tree C++/HIP extensions built by setup.py / __graft_entry__.build()
for gfx950.  Import errors propagate loudly on GPU boxes — there is no silent
eager fallback for the compute path.
"""
