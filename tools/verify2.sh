#!/bin/bash
set -u
cd /root/repo
R=gpurun_out/final2; mkdir -p $R
echo "=== soak 300 steps (lr 0.02 default) ===" | tee $R/summary.log
timeout 400 python bench.py --gpus 1 --steps 300 --warmup 10 > $R/soak.json 2>$R/soak.err
tail -1 $R/soak.json | tee -a $R/summary.log
echo "=== resnet50 bench ===" | tee -a $R/summary.log
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 --model resnet50 > $R/resnet.json 2>$R/resnet.err
tail -1 $R/resnet.json | tee -a $R/summary.log
echo "=== 2-rank 40 steps ===" | tee -a $R/summary.log
timeout 300 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 --master-addr 127.0.0.1 --master-port 29741 bench.py --gpus 2 --steps 40 --warmup 10 > $R/r2.json 2>$R/r2.err
grep -h '"metric"' $R/r2.json | tail -1 | tee -a $R/summary.log
echo "=== mnist headline re-check ===" | tee -a $R/summary.log
timeout 240 python bench.py --gpus 1 --steps 60 --warmup 12 > $R/mnist.json 2>$R/mnist.err
tail -1 $R/mnist.json | tee -a $R/summary.log
echo DONE | tee -a $R/summary.log
