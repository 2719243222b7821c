"""Phase-level corruption tracker: train_step -> allreduce -> opt, 8 iters."""
import os, struct, torch
from learningorchestra_amd.parallel import (all_reduce_grads, get_rank,
                                            get_world_size, init_distributed)
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

local = init_distributed()
rank, world = get_rank(), get_world_size()
dev = f"cuda:{local % torch.cuda.device_count()}"
torch.cuda.set_device(dev)
m = build_mnist_cnn(dev, seed=1)
tr = Trainer(m, make_sgd(m, lr=0.02), device=dev, use_graph=False)
b = 4096
x, y = mnist_batch(b, device=dev, dtype=torch.bfloat16, seed=1234 + rank)
a = m.arena
gscale = 1.0 / (b * world)

def check(tag, it):
    torch.cuda.synchronize()
    g = a.grad
    bad = (g.abs() > 1e3) | torch.isnan(g) | torch.isinf(g)
    n = int(bad.sum())
    if n:
        idx = bad.nonzero().flatten()[:4].cpu()
        vals = g[idx].cpu()
        bits = [struct.pack("f", float(v)).hex() for v in vals]
        names = []
        for i in idx.tolist():
            for nm, (o, s) in a._offsets.items():
                if o <= i < o + s:
                    names.append(f"{nm}+{i-o}")
                    break
        print(f"rank{rank} it{it} [{tag}] nbad={n} first={list(zip(names, bits))}",
              flush=True)
        return True
    return False

hit = False
for it in range(8):
    m.train_step(x, y, gscale=gscale)
    hit = check("post-train", it) or hit
    all_reduce_grads(a.grad)
    hit = check("post-allreduce", it) or hit
    tr._opt_body()
    hit = check("post-opt", it) or hit
    if hit:
        break
if not hit and rank == 0:
    print("CLEAN 8 iters", flush=True)
