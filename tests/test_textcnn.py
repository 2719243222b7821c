"""TextCNN (IMDb config) on the engine — CPU reference path."""
import torch

from learningorchestra_amd.data.synthetic import imdb_batch
from learningorchestra_amd.engine.trainer import Trainer, make_sgd
from learningorchestra_amd.models.textcnn import build_textcnn


def test_textcnn_step_shapes():
    m = build_textcnn("cpu", seed=0, vocab=500, emb_dim=16, filters=8)
    ids, y = imdb_batch(8, seq_len=32, vocab=500, seed=0)
    loss, correct = m.train_step(ids, y)
    assert loss.item() > 0 and 0 <= correct.item() <= 8
    preds = m.predict(ids)
    assert preds.shape == (8,) and preds.max() < 2


def test_textcnn_learns():
    torch.manual_seed(0)
    m = build_textcnn("cpu", seed=1, vocab=100, emb_dim=16, filters=16,
                      kernel_sizes=(3,))
    trainer = Trainer(m, make_sgd(m, lr=0.1), device="cpu")

    def batch():
        # token 7 present in first half => label 1
        y = torch.randint(0, 2, (32,))
        ids = torch.randint(10, 100, (32, 24))
        ids[y == 1, 3] = 7
        return ids, y

    first, _ = trainer.step(*batch())
    for _ in range(60):
        trainer.step_async(*batch())
    last, acc = trainer.step(*batch())
    assert last < first * 0.7, (first, last)
    assert acc > 0.7


def test_embedding_grads():
    from learningorchestra_amd.engine.arena import ParamArena
    from learningorchestra_amd.engine.layers import Embedding
    lay = Embedding("e", 10, 8)
    arena = ParamArena("cpu")
    lay.build(arena)
    arena.finalize()
    ids = torch.tensor([[1, 2], [1, 3]])
    out = lay.forward(ids)
    assert out.shape == (2, 2, 8)
    assert torch.allclose(out[0, 0].float(), arena.p("e.w")[1].float())
    dy = torch.ones(2, 2, 8).bfloat16()
    lay.backward(dy)
    g = arena.g("e.w")
    assert torch.allclose(g[1], torch.full((8,), 2.0))  # id 1 hit twice
    assert torch.allclose(g[2], torch.full((8,), 1.0))
    assert g[0].abs().max() == 0
