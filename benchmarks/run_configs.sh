#!/bin/bash
# Re-emit the per-config benchmark records with round-2 kernels and the
# fixed per-config metric labels.  Run on an MI355X box.
set -x
cd "$(dirname "$0")/.."
mkdir -p benchmarks/results gpurun_out

run() {  # name, args...
  local name=$1; shift
  timeout 900 python bench.py "$@" 2>gpurun_out/cfg_${name}.err \
    | tail -1 > benchmarks/results/r02_bench_${name}.json
  tail -3 gpurun_out/cfg_${name}.err
  cat benchmarks/results/r02_bench_${name}.json
}

run llama8b_dp1        --steps 4 --warmup 2
run llama8b_mxfp8      --steps 3 --warmup 1 --quant mxfp8
run llama8b_fp8        --steps 3 --warmup 1 --quant fp8
run mixtral_dp1        --steps 3 --warmup 1 --model mixtral-8x7b
run llama70b_fp8_beam8 --steps 2 --warmup 1 --model llama-3-70b --quant fp8 --beam-width 8 --branch-factor 4
