#!/bin/bash
# Round-3 kernel validation pass — run FIRST THING on the GPU box:
#
#   gpurun --timeout 1500 -- 'bash scripts/validate_gated_kernels.sh \
#       > gpurun_out/gated_validation.log 2>&1'
#
# Validates every kernel variant written+compile-checked (but not GPU-run)
# in late round 2, in dependency order, with per-group timeouts so one bad
# kernel cannot strike the box. On success, flip the gates' defaults
# (GPUSTACK_AMD_OSS_KERNELS / GPUSTACK_AMD_MLA_KERNEL in
# gpustack_amd/ops/__init__.py + models/llama.py + engine/model_runner.py)
# and re-run the FULL `pytest -m gpu` suite before relying on them.
set -x
cd "$(dirname "$0")/.."

echo "=== 1. validated baseline still green (fast sanity) ==="
timeout 300 python -m pytest tests/test_ops_gpu.py -q -m gpu \
    -k "not oss and not mla" -x || exit 1

echo "=== 2. GPT-OSS attention variants (D64 + sinks + window) ==="
GPUSTACK_AMD_OSS_KERNELS=1 timeout 420 python -m pytest \
    tests/test_ops_gpu.py -q -m gpu -k "oss_paged_attn or oss_varlen or oss_paged_prefill" || exit 1

echo "=== 3. Gemma variants (D256 + softcap) ==="
GPUSTACK_AMD_OSS_KERNELS=1 timeout 300 python -m pytest \
    tests/test_ops_gpu.py -q -m gpu -k "softcap_d256" || exit 1

echo "=== 4. fused MoE clamped-swiglu + biases ==="
GPUSTACK_AMD_OSS_KERNELS=1 timeout 180 python -m pytest \
    tests/test_ops_gpu.py -q -m gpu -k "oss_fused_moe" || exit 1

echo "=== 5. MLA absorbed decode + expand prefill (192/128) ==="
GPUSTACK_AMD_MLA_KERNEL=1 timeout 300 python -m pytest \
    tests/test_ops_gpu.py -q -m gpu -k "mla" || exit 1

echo "=== 6. kernel micro-bench (commit summaries to profiles/) ==="
GPUSTACK_AMD_MLA_KERNEL=1 timeout 240 python scripts/bench_mla.py \
    | tee gpurun_out/mla_bench.txt

echo "=== 7. engine-level serving on the gated shapes ==="
GPUSTACK_AMD_OSS_KERNELS=1 timeout 420 python -m pytest \
    tests/test_engine_gpu.py -q -m gpu -k "oss_shape or gemma_shape" || exit 1
GPUSTACK_AMD_MLA_KERNEL=1 timeout 300 python -m pytest \
    tests/test_engine_gpu.py -q -m gpu -k "mla_shape" || exit 1

echo "ALL GATED KERNEL GROUPS VALIDATED"
echo "next: flip gate defaults, run full 'pytest -m gpu', then bench"
echo "  gpt-oss-20b / gemma-2-9b / deepseek CPU-vs-GPU logits spot-checks"
