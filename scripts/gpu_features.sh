#!/bin/bash
# First-GPU-window plan for the features added after this round's budget
# was spent (run via gpurun; outputs under gpurun_out/):
#   1. late-file GPU tests (spec decode, qwen2, gpt2, extended GQA,
#      prefix caching) -- the correctness gate
#   2. llm_features_bench on llama-3-8b -- wall-time effect of prefix
#      caching + speculation at production scale
#   3. TP A/B when an 8-GPU window exists: CMLS_TP_MICROBATCH=1 vs decode
#      graphs (torchrun -- see docs/ROADMAP.md TP section)
set -x
mkdir -p gpurun_out
python -m pytest tests/test_zz_late_gpu.py -q -m gpu \
    > gpurun_out/features_tests.log 2>&1
python benchmarks/llm_features_bench.py --preset llama-3-8b \
    --requests 16 --max-tokens 128 \
    > gpurun_out/features_bench.json 2> gpurun_out/features_bench.err
tail -5 gpurun_out/features_tests.log
cat gpurun_out/features_bench.json
