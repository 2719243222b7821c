"""world=2 on one GPU: (1) all_reduce of known pattern, (2) train_step with
no all-reduce, (3) train_step + all_reduce with explicit pre-sync."""
import os, torch
import torch.distributed as dist
from learningorchestra_amd.parallel import (all_reduce_grads, get_rank,
                                            get_world_size, init_distributed)
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

local = init_distributed()
rank, world = get_rank(), get_world_size()
dev = f"cuda:{local % torch.cuda.device_count()}"
torch.cuda.set_device(dev)

# (1) pure all-reduce of ones
for sz in (318832, 1 << 20, 1 << 24):
    bad = 0
    for it in range(10):
        t = torch.ones(sz, device=dev)
        all_reduce_grads(t)
        wrong = int((t != float(world)).sum())
        bad += 1 if wrong else 0
    if rank == 0:
        print(f"[allreduce ones sz={sz}] bad_iters {bad}/10", flush=True)

# (2) train_step only, no all-reduce
m = build_mnist_cnn(dev, seed=0)
tr = Trainer(m, make_sgd(m, lr=0.02), device=dev, use_graph=False)
b = 4096
x, y = mnist_batch(b, device=dev, dtype=torch.bfloat16, seed=1234 + rank)
a = m.arena
gscale = 1.0 / (b * world)
def bad_params():
    torch.cuda.synchronize()
    out = []
    for n, (o, s) in sorted(a._offsets.items(), key=lambda kv: kv[1][0]):
        g = float(a.grad[o:o + s].abs().max())
        if g > 1e3 or g != g:
            out.append(f"{n}:{g:.2e}")
    return out
for i in range(5):
    m.train_step(x, y, gscale=gscale)
    tr._opt_body()
print(f"[rank{rank} no-allreduce x5] bad {bad_params()} |w|max {float(a.master.abs().max()):.3e}", flush=True)

# (3) train_step + all_reduce with explicit sync before staging
m2 = build_mnist_cnn(dev, seed=1)
tr2 = Trainer(m2, make_sgd(m2, lr=0.02), device=dev, use_graph=False)
a2 = m2.arena
for i in range(5):
    m2.train_step(x, y, gscale=gscale)
    torch.cuda.synchronize()
    all_reduce_grads(a2.grad)
    tr2._opt_body()
torch.cuda.synchronize()
bad2 = []
for n, (o, s) in sorted(a2._offsets.items(), key=lambda kv: kv[1][0]):
    g = float(a2.grad[o:o + s].abs().max())
    if g > 1e3 or g != g:
        bad2.append(f"{n}:{g:.2e}")
print(f"[rank{rank} presync-allreduce x5] bad {bad2} |w|max {float(a2.master.abs().max()):.3e}", flush=True)
