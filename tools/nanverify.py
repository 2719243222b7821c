"""Post-fix verification: trainer path 30 iters, eager + graph, 2 ranks."""
import os, torch
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
tr = Trainer(m, make_sgd(m, lr=0.02), device=dev, use_graph=use_graph)
b = int(os.environ.get("BB", "4096"))
x, y = mnist_batch(b, device=dev, dtype=torch.bfloat16, seed=1234 + rank)
a = m.arena
nbad = 0
first = -1
for i in range(30):
    tr.step_async(x, y)
    torch.cuda.synchronize()
    g = a.grad
    if bool(((g.abs() > 1e3) | torch.isnan(g)).any()):
        nbad += 1
        if first < 0:
            first = i
            per = []
            for n, (o, sz) in a._offsets.items():
                gg = g[o:o + sz]
                if bool(((gg.abs() > 1e3) | torch.isnan(gg)).any()):
                    per.append(n)
            print(f"rank{rank} FIRST bad at it{i} params={per}", flush=True)
wbad = bool((torch.isnan(a.master) | (a.master.abs() > 1e3)).any())
print(f"rank{rank} graph={use_graph}: {nbad}/30 bad grad iters, master_bad={wbad}, "
      f"loss {float(m.loss_sum)/b:.4f}", flush=True)
