#!/bin/bash
# Host-side AddressSanitizer pass over the HIP extension (SURVEY.md §5
# race-detection/sanitizer obligation; VERDICT r1 gap A2).
#
# Builds a separate ASan instrumented .so (HOST code only — device code is
# unaffected; kernel-side memory correctness is covered by the fault-free
# parity suite + the guarded-store audits) and runs the GPU op tests under
# it. Catches host-side heap misuse in the bindings, the hipBLASLt plan
# cache, the workspace management and the launcher glue.
#
# Usage (on a GPU box):  bash scripts/sanitize.sh [pytest-args...]
set -e
cd "$(dirname "$0")/.."

export VILBERT_ASAN_BUILD=1
python -m vilbert_multi_task_amd.ops.build --force --asan

# hipcc instruments with CLANG's ASan runtime — preload that one, not gcc's
ASAN_LIB=$(ls /opt/rocm/lib/llvm/lib/clang/*/lib/linux/libclang_rt.asan-x86_64.so | head -1)
echo "== running GPU op tests under host ASan ($ASAN_LIB) =="
LD_PRELOAD="$ASAN_LIB" \
ASAN_OPTIONS=detect_leaks=0:protect_shadow_gap=0:replace_intrin=0:alloc_dealloc_mismatch=0:allocator_may_return_null=1 \
python -m pytest tests/test_gpu_ops.py -q "${@:--x}"
