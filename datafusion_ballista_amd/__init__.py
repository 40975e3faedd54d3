"""datafusion_ballista_amd — MI355X-native stage executor for Ballista.

Python host bindings over the C ABI in include/ballista_gpu.h
(libballista_gpu.so: hand-written HIP/CDNA4 kernels for gfx950).  This is
the executor-side hot path of SURVEY.md §8 — FilterExec predicate eval,
create_hashes/hash-repartition, stable multi-split, gather, and the fused
TPC-H q6/q1 filter+aggregate stages — behind the C-ABI seam a Rust
`GpuExecutionEngine` (Ballista's `ExecutionEngine` trait,
ballista/executor/src/execution_engine.rs:53-103) would bind; see
INTEGRATION.md for the Rust-side stub.

There is NO CPU fallback: on a machine with a GPU, every op runs through
libballista_gpu.so or raises.  The CPU oracle under oracle/ is test
infrastructure only and is never imported here.
"""

from .gpu import (  # noqa: F401
    BG_DT_INT32, BG_DT_INT64, BG_DT_DATE32, BG_DT_DECIMAL128, BG_DT_DICT8,
    BG_PRED_GE_LT, BG_PRED_BETWEEN, BG_PRED_LT, BG_PRED_EQ, BG_PRED_GT,
    GpuStageContext, lib_path, load_library,
)
