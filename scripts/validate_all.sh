#!/bin/bash
# Round-2 full validation battery (run on a GPU box).
set -u
cd "$(dirname "$0")/.."
echo "=== pytest gpu (x2) ==="
for i in 1 2; do
  timeout 420 python -m pytest tests -m gpu -q > gpurun_out/val_suite$i.log 2>&1
  echo "suite$i rc=$?: $(grep -E 'passed|failed' gpurun_out/val_suite$i.log | tail -1)"
done
echo "=== smoke ==="
timeout 180 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
echo "=== bench train (20 steps) ==="
timeout 240 python bench.py --steps 20 --warmup 5 2>/dev/null | tail -1
echo "=== bench infer ==="
timeout 240 python bench.py --steps 15 --warmup 5 --mode infer 2>/dev/null | tail -1 | python3 -c "import json,sys; print(round(json.load(sys.stdin)['value'],1),'img/s infer')"
echo "=== torchrun world=1 ==="
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29887 bench.py --gpus 1 --steps 12 --warmup 4 2>/dev/null | tail -1 | python3 -c "import json,sys; print(round(json.load(sys.stdin)['value'],1),'img/s torchrun')"
echo "=== stretch + video ==="
timeout 420 python scripts/config_bench.py 2>/dev/null
echo "=== soak 300 ==="
timeout 600 python scripts/soak.py 2>/dev/null | tail -2
echo "=== examples (GPU) ==="
for ex in examples/01_inference.py examples/03_stateful_video.py examples/04_trajectory_loss.py; do
  timeout 180 python $ex > /dev/null 2>&1 && echo "$ex OK" || echo "$ex FAILED"
done
timeout 240 python examples/02_denoising_training.py > /dev/null 2>&1 && echo "examples/02 OK" || echo "examples/02 FAILED"
echo "=== done ==="
