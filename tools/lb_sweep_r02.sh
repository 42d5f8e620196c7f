#!/bin/bash
# LB strategy sweep on hub-heavy shapes
for shape in "21297772 530051090 twitter" "2997166 212698418 orkut"; do
  set -- $shape
  for lb in cm strict none; do
    echo "== $3 lb=$lb =="
    GRAPEHIP_LB=$lb python tools/time_apps.py --nv $1 --ne $2 \
      --apps bfs,sssp 2>/dev/null | tail -1
  done
done
