"""Does the gloo call matter, or just the D2H/H2D staging copies?
Variants via env V: A=pageable copies no-gloo, B=pinned copies no-gloo,
C=gloo allreduce with pinned staging, D=gloo allreduce pageable (baseline)."""
import os, torch
import torch.distributed as dist
from learningorchestra_amd.parallel import (get_rank, get_world_size,
                                            init_distributed)
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

V = os.environ.get("V", "D")
LOCK = os.environ.get("LOCK", "0") == "1"
import fcntl
_lockf = open("/tmp/lo_gpu_lock", "w") if LOCK else None
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
pinned = torch.empty(a.grad.numel(), dtype=a.grad.dtype, pin_memory=True)
nbad = 0
for it in range(15):
    if _lockf:
        fcntl.flock(_lockf, fcntl.LOCK_EX)
    m.train_step(x, y, gscale=gscale)
    if _lockf:
        torch.cuda.synchronize()
        fcntl.flock(_lockf, fcntl.LOCK_UN)
    g = a.grad
    if V == "E":
        torch.cuda.synchronize()
        host = g.detach().to("cpu")
        dist.all_reduce(host, op=dist.ReduceOp.SUM)
        g.copy_(host)
        torch.cuda.synchronize()
    elif V == "A":
        host = g.detach().to("cpu")
        g.copy_(host)
    elif V == "B":
        pinned.copy_(g, non_blocking=False)
        g.copy_(pinned, non_blocking=False)
    elif V == "C":
        pinned.copy_(g, non_blocking=False)
        dist.all_reduce(pinned, op=dist.ReduceOp.SUM)
        g.copy_(pinned, non_blocking=False)
    else:
        host = g.detach().to("cpu")
        dist.all_reduce(host, op=dist.ReduceOp.SUM)
        g.copy_(host)
    torch.cuda.synchronize()
    bad = (g.abs() > 1e3) | torch.isnan(g)
    if bad.any():
        nbad += 1
    tr._opt_body()
print(f"V={V} rank{rank}: {nbad}/15 bad iters", flush=True)
