"""Pure-PyTorch 2-rank oversubscription probe: same loop shape as nanrepro7
variant E (max rank alignment), zero learningorchestra kernels."""
import os, torch
import torch.nn as tnn
import torch.distributed as dist

dist.init_process_group("gloo")
rank = dist.get_rank()
torch.cuda.set_device(0)
dev = "cuda:0"
torch.manual_seed(1)
net = tnn.Sequential(
    tnn.Conv2d(1, 32, 5), tnn.ReLU(), tnn.MaxPool2d(2),
    tnn.Conv2d(32, 64, 5), tnn.ReLU(), tnn.MaxPool2d(2),
    tnn.Flatten(), tnn.Linear(64 * 16, 256), tnn.ReLU(),
    tnn.Linear(256, 10)).to(dev).to(torch.bfloat16)
opt = torch.optim.SGD(net.parameters(), lr=0.02, momentum=0.9)
g = torch.Generator().manual_seed(1234 + rank)
x = torch.randn(4096, 1, 28, 28, generator=g).to(dev).to(torch.bfloat16)
y = torch.randint(0, 10, (4096,), generator=g).to(dev)
lossf = tnn.CrossEntropyLoss()
nbad = 0
for it in range(15):
    opt.zero_grad(set_to_none=False)
    loss = lossf(net(x).float(), y)
    loss.backward()
    torch.cuda.synchronize()
    flat = torch.cat([p.grad.reshape(-1).float() for p in net.parameters()])
    host = flat.to("cpu")
    dist.all_reduce(host, op=dist.ReduceOp.SUM)
    # write back (splice) to mimic the same H2D traffic
    dev_sum = host.to(dev)
    off = 0
    for p in net.parameters():
        n = p.grad.numel()
        p.grad.copy_(dev_sum[off:off + n].reshape(p.grad.shape)
                     .to(p.grad.dtype) * 0.5)
        off += n
    torch.cuda.synchronize()
    bad = (flat.abs() > 1e3) | torch.isnan(flat)
    bad2 = False
    for p in net.parameters():
        if bool(torch.isnan(p.grad).any() | (p.grad.abs() > 1e3).any()):
            bad2 = True
    if bad.any() or bad2:
        nbad += 1
    opt.step()
print(f"torch-only rank{rank}: {nbad}/15 bad iters, last loss {float(loss):.4f}", flush=True)
