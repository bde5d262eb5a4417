#!/bin/bash
# First-GPU-hour validation sweep (ROADMAP item 7): run the whole GPU
# tier, the serving load benchmark with every round-1 serving feature
# on, the big-model smokes, and a headline bench point. Run via:
#   /usr/local/graft/bin/gpurun --timeout 3000 -- 'bash scripts/gpu_validate.sh'
set -x
mkdir -p gpurun_out
python -m pytest tests -m gpu -q -x 2>&1 | tail -5 | tee gpurun_out/val_pytest.log

python scripts/bench_serving.py --model Qwen/Qwen2.5-7B-Instruct \
    --rate 8 --num-requests 64 --prompt-len 1024 --new-tokens 128 \
    --max-slots 32 --max-ctx 4096 --prefill-chunk 512 \
    --prefix-caching --shared-prefix 512 \
    > gpurun_out/val_serving.json 2>gpurun_out/val_serving.err
tail -1 gpurun_out/val_serving.json

timeout 900 python scripts/smoke_big_models.py all \
    > gpurun_out/val_bigmodels.log 2>&1
tail -8 gpurun_out/val_bigmodels.log

python bench.py --gpus 1 --steps 3 --warmup 1 \
    > gpurun_out/val_bench.json 2>gpurun_out/val_bench.err
tail -1 gpurun_out/val_bench.json
