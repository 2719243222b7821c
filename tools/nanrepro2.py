"""Split-graph corruption bisect: warmup / captureA / captureB / replay."""
import os, torch
from learningorchestra_amd.parallel import (all_reduce_grads, get_rank,
                                            get_world_size, init_distributed)
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

local = init_distributed()
rank, world = get_rank(), get_world_size()
dev = f"cuda:{local % torch.cuda.device_count()}"
torch.cuda.set_device(dev)
m = build_mnist_cnn(dev, seed=0)
tr = Trainer(m, make_sgd(m, lr=0.02), device=dev, use_graph=False)
b = 4096
x, y = mnist_batch(b, device=dev, dtype=torch.bfloat16, seed=1234 + rank)
a = m.arena

def report(tag):
    torch.cuda.synchronize()
    gm, wm = float(a.grad.abs().max()), float(a.master.abs().max())
    per = []
    for n, (o, s) in sorted(a._offsets.items(), key=lambda kv: kv[1][0]):
        g = float(a.grad[o:o + s].abs().max())
        if g > 1e3 or g != g:
            per.append(f"{n}:{g:.2e}")
    if rank == 0:
        print(f"[{tag}] |g|max {gm:.3e} |w|max {wm:.3e} bad:[{' '.join(per)}]",
              flush=True)

gscale = 1.0 / (b * world)
for i in range(3):
    m.train_step(x, y, gscale=gscale)
    all_reduce_grads(a.grad)
    tr._opt_body()
report("warmup3")

torch.cuda.synchronize()
gA = torch.cuda.CUDAGraph()
with torch.cuda.graph(gA, capture_error_mode="thread_local"):
    m.train_step(x, y, gscale=gscale)
report("captureA")
gB = torch.cuda.CUDAGraph()
with torch.cuda.graph(gB, capture_error_mode="thread_local"):
    tr._opt_body()
report("captureB")
for i in range(3):
    gA.replay()
    report(f"replayA{i}")
    all_reduce_grads(a.grad)
    report(f"allreduce{i}")
    gB.replay()
    report(f"replayB{i}")
