#!/bin/bash
# Final round-2 validation marathon. Run on the GPU box via gpurun.
set -u
cd /root/repo
mkdir -p gpurun_out
R=gpurun_out/final4
mkdir -p $R
echo "=== freshness ===" | tee $R/summary.log
SO=$(ls learningorchestra_amd/_build/*.so 2>/dev/null | head -1)
ls -la learningorchestra_amd/_build/ 2>/dev/null | tail -3 | tee -a $R/summary.log
python - << 'PY' 2>&1 | tee -a $R/summary.log
import learningorchestra_amd.ops.functional as F
import torch
print("ext loaded:", F._ext is not None if hasattr(F, "_ext") else "n/a")
print("device:", torch.cuda.get_device_name(0))
PY

echo "=== pytest -m gpu (full) ===" | tee -a $R/summary.log
timeout 600 python -m pytest tests -m gpu -x -q 2>&1 | tail -5 | tee -a $R/summary.log

echo "=== smoke ===" | tee -a $R/summary.log
timeout 180 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" 2>&1 | tail -3 | tee -a $R/summary.log

echo "=== fuzz 3 seeds ===" | tee -a $R/summary.log
for s in 101 202 303; do
  timeout 200 python tools/conv_fuzz.py 40 $s 2>&1 | tail -1 | tee -a $R/summary.log
done

echo "=== gemm256 raceprobe (tile kernels must be clean) ===" | tee -a $R/summary.log
timeout 240 python tools/gemm256_raceprobe.py 2>&1 | tail -3 | tee -a $R/summary.log

echo "=== bench: mnist x2 ===" | tee -a $R/summary.log
timeout 240 python bench.py --gpus 1 --steps 60 --warmup 12 > $R/bench_mnist_1.json 2>$R/bench_mnist_1.err
tail -1 $R/bench_mnist_1.json | tee -a $R/summary.log
timeout 240 python bench.py --gpus 1 --steps 60 --warmup 12 > $R/bench_mnist_2.json 2>$R/bench_mnist_2.err
tail -1 $R/bench_mnist_2.json | tee -a $R/summary.log

echo "=== bench: textcnn ===" | tee -a $R/summary.log
timeout 240 python bench.py --gpus 1 --steps 40 --warmup 10 --model textcnn > $R/bench_textcnn.json 2>$R/bench_textcnn.err
tail -1 $R/bench_textcnn.json | tee -a $R/summary.log

echo "=== bench: resnet ===" | tee -a $R/summary.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --model resnet50 > $R/bench_resnet.json 2>$R/bench_resnet.err
tail -1 $R/bench_resnet.json | tee -a $R/summary.log

echo "=== bench: gbt ===" | tee -a $R/summary.log
timeout 240 python bench.py --gpus 1 --steps 30 --warmup 5 --model gbt > $R/bench_gbt.json 2>$R/bench_gbt.err
tail -1 $R/bench_gbt.json | tee -a $R/summary.log

echo "=== bench: 2 ranks on 1 GPU (RCCL) ===" | tee -a $R/summary.log
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29731 bench.py --gpus 2 --steps 40 --warmup 10 > $R/bench_2rank.json 2>$R/bench_2rank.err
grep -h '"metric"' $R/bench_2rank.json | tail -1 | tee -a $R/summary.log

echo "=== soak 300 steps ===" | tee -a $R/summary.log
timeout 400 python bench.py --gpus 1 --steps 300 --warmup 10 > $R/soak.json 2>$R/soak.err
tail -1 $R/soak.json | tee -a $R/summary.log

echo "=== rocprof kernel stats (mnist) ===" | tee -a $R/summary.log
export TMPDIR=/tmp
(cd /tmp && timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/final/prof -o mnist -- python /root/repo/bench.py --gpus 1 --steps 20 --warmup 5 > /root/repo/$R/prof_mnist.log 2>&1)
find $R/prof -name '*stats*' | tee -a $R/summary.log

echo "=== DONE ===" | tee -a $R/summary.log
