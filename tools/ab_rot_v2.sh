#!/usr/bin/env bash
# A/B the RELORA_AMD_ROT_V2 kernel build (bank-model-verified conflict-free
# LDS layouts) against the default rotation on a GPU box, in one run:
#
#   /usr/local/graft/bin/gpurun --timeout 1800 -- 'bash tools/ab_rot_v2.sh > gpurun_out/ab_rot_v2.log 2>&1'
#
# Rebuild is per-variant (only attention.hip/lora_gemm.hip recompile).
set -euo pipefail
cd "$(dirname "$0")/.."

STEPS=${STEPS:-8}
WARMUP=${WARMUP:-3}

run_variant() {
    local name="$1"; shift
    local flag="$1"; shift
    echo "=== building variant: $name ==="
    touch relora_amd/ops/csrc/attention.hip relora_amd/ops/csrc/lora_gemm.hip
    env RELORA_AMD_ROT_V2="$flag" PYTORCH_ROCM_ARCH=gfx950 \
        python setup.py build_ext --inplace > /dev/null
    echo "=== $name: attention microbench ==="
    python tools/bench_attention.py || true
    echo "=== $name: lora microbench ==="
    python tools/bench_lora.py || true
    echo "=== $name: e2e bench ==="
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP"
}

run_variant "default" ""
run_variant "rot_v2" "1"

# leave the default build in place
touch relora_amd/ops/csrc/attention.hip relora_amd/ops/csrc/lora_gemm.hip
env PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace > /dev/null
echo "=== done (default build restored) ==="
