#!/bin/bash
# world-2 bench plumbing smoke on ONE GPU (TCP data plane, small shape)
for r in 0 1; do
  RANK=$r LOCAL_RANK=$r WORLD_SIZE=2 MASTER_ADDR=127.0.0.1 MASTER_PORT=29930 \
    python bench.py --gpus 2 --steps 1 --warmup 0 --nv 4000000 --ne 40000000 &
done
wait
