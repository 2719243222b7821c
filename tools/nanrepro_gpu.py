"""2-rank NaN hunt: loss trajectory with/without split-graph mode."""
import os, sys, torch
from learningorchestra_amd.parallel import get_rank, init_distributed
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

use_graph = os.environ.get("NG", "0") != "1"
local = init_distributed()
rank = get_rank()
dev = f"cuda:{local % torch.cuda.device_count()}"
torch.cuda.set_device(dev)
m = build_mnist_cnn(dev, seed=0)
if os.environ.get("MC", "1") == "0":   # disable masked-colsum bias path
    for lay in m.layers:
        if hasattr(lay, "_fused_pool"):
            lay._fused_pool = None
tr = Trainer(m, make_sgd(m, lr=0.02), device=dev, use_graph=use_graph)
b = 32768
x, y = mnist_batch(b, device=dev, dtype=torch.bfloat16, seed=1234 + rank)
for i in range(12):
    tr.step_async(x, y)
    if i % 5 == 4 or i < 3:
        torch.cuda.synchronize()
        ls = float(m.loss_sum) / b
        gmax = float(m.arena.grad.abs().max())
        wmax = float(m.arena.master.abs().max())
        print(f"rank{rank} it{i+1} loss {ls:.4f} |g|max {gmax:.3e} |w|max {wmax:.3e}",
              flush=True)
