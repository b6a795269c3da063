#!/bin/bash
# CI-optional kernel-sanity lane: run the GPU test suite
# with the ROCm debug/serialization knobs that surface race conditions,
# OOB accesses and queue corruption which a normal run can hide.
#   AMD_SERIALIZE_KERNEL=3  — serialize launches + sync after each kernel
#   AMD_SERIALIZE_COPY=3    — same for copies
#   HSA_SVM_GUARD_PAGES=1   — guard pages around SVM allocations
#   PYTORCH_NO_HIP_MEMORY_CACHING=1 — every alloc hits the driver, so
#                              use-after-free faults instead of recycling
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
export AMD_SERIALIZE_KERNEL=3 AMD_SERIALIZE_COPY=3
export HSA_SVM_GUARD_PAGES=1 PYTORCH_NO_HIP_MEMORY_CACHING=1
timeout 1800 python -m pytest tests/test_ops_gpu.py -m gpu -x -q "$@"
