#!/bin/bash
for r in 0 1 2 3; do
  RANK=$r LOCAL_RANK=$r WORLD_SIZE=4 MASTER_ADDR=127.0.0.1 MASTER_PORT=29934 \
    python bench.py --gpus 4 --steps 1 --warmup 0 --nv 4000000 --ne 40000000 &
done
wait
