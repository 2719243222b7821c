"""Single process train_step loop with bad-grad detection (run vs hammer)."""
import sys, torch
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
from learningorchestra_amd.data.synthetic import mnist_batch

b = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 60
m = build_mnist_cnn("cuda:0", seed=1)
tr = Trainer(m, make_sgd(m, lr=0.02), device="cuda:0", use_graph=False)
x, y = mnist_batch(b, device="cuda:0", dtype=torch.bfloat16, seed=1234)
a = m.arena
gscale = 1.0 / b
nbad = 0
for it in range(iters):
    m.train_step(x, y, gscale=gscale)
    torch.cuda.synchronize()
    g = a.grad
    bad = (g.abs() > 1e3) | torch.isnan(g)
    if bad.any():
        nbad += 1
        per = []
        for n, (o, s) in a._offsets.items():
            nb = int(bad[o:o + s].sum())
            if nb:
                per.append(f"{n}:{nb}")
        print(f"it{it} BAD {per}", flush=True)
        if nbad >= 5: break
    tr._opt_body()
print(f"done: {nbad}/{iters} bad iters (b={b})", flush=True)
