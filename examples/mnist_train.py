"""Direct engine usage — training the flagship MNIST-CNN without the REST
layer: build model -> Trainer (hipGraph-captured on GPU) -> train ->
checkpoint -> resume.

Run:  python examples/mnist_train.py [steps]
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from learningorchestra_amd.data.synthetic import mnist_batch
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 30
device = "cuda" if torch.cuda.is_available() else "cpu"
batch = 4096 if device == "cuda" else 128

model = build_mnist_cnn(device, seed=0)
trainer = Trainer(model, make_sgd(model, lr=0.05), device=device,
                  use_graph=(device == "cuda"))

def data():
    while True:
        yield mnist_batch(batch, device=device, dtype=torch.bfloat16,
                          seed=torch.randint(0, 1 << 30, ()).item())

stats = trainer.train(data(), steps=steps, log_every=max(1, steps // 3),
                      log_fn=print)
print(f"{stats['samples_per_sec']:.0f} samples/s over {steps} steps "
      f"on {device}")

ck = os.path.join(tempfile.gettempdir(), "mnist_ck.pt")
trainer.save_checkpoint(ck, step=steps)
resumed = trainer.load_checkpoint(ck)
print(f"checkpoint round-trip OK (resumed at step {resumed})")
