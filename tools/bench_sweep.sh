#!/bin/bash
# Round-end bench sweep: all modes at defaults + headline batch sizes.
set -e
for m in mlp sqlagg proto_mlp bert; do
  echo "== $m (default batch)"
  timeout 150 python bench.py --mode $m --steps 40 --warmup 8 2>/dev/null | tail -1
done
echo "== mlp @65536"
timeout 150 python bench.py --batch 65536 --steps 30 --warmup 5 2>/dev/null | tail -1
echo "== sqlagg @1048576"
timeout 150 python bench.py --mode sqlagg --batch 1048576 --steps 20 --warmup 4 2>/dev/null | tail -1
echo "== proto_mlp @262144"
timeout 150 python bench.py --mode proto_mlp --batch 262144 --steps 20 --warmup 4 2>/dev/null | tail -1
