"""Single-process eager corruption probe across batch sizes."""
import sys, torch
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

for b in (512, 1024, 2048, 4096, 8192, 32768):
    m = build_mnist_cnn("cuda:0", seed=0)
    tr = Trainer(m, make_sgd(m, lr=0.02), device="cuda:0", use_graph=False)
    x, y = mnist_batch(b, device="cuda:0", dtype=torch.bfloat16, seed=1234)
    a = m.arena
    for i in range(3):
        tr.step_async(x, y)
    torch.cuda.synchronize()
    per = []
    for n, (o, s) in sorted(a._offsets.items(), key=lambda kv: kv[1][0]):
        g = float(a.grad[o:o + s].abs().max())
        if g > 1e3 or g != g:
            per.append(f"{n}:{g:.2e}")
    print(f"b={b} loss {float(m.loss_sum)/b:.4f} |g|max {float(a.grad.abs().max()):.3e} "
          f"|w|max {float(a.master.abs().max()):.3e} bad:[{' '.join(per)}]",
          flush=True)
