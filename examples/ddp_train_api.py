"""Multi-GPU training through the REST API (round-2 capability).

The reference fanned builder jobs out to 3 Spark workers; here a train POST
with ``"gpus": N`` spawns an N-rank torchrun job — one process per GPU over
RCCL — from the API server's scheduler. Runs against the embedded runtime
(no server process needed); point Context at a host:port for a live server.

    python examples/ddp_train_api.py [n_gpus]
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from learning_orchestra_client import Context  # noqa: E402
from learningorchestra_amd.api.app import PREFIX, Runtime  # noqa: E402


def main() -> None:
    gpus = int(sys.argv[1]) if len(sys.argv) > 1 else 2
    ctx = Context.in_process(Runtime())
    http = ctx._session  # TestClient-compatible session

    # idempotent re-runs: clear any previous artifacts of the same names
    for verb, name in (("train", "ddp_fit"), ("model", "ddp_cnn")):
        http.delete(f"{PREFIX}/{verb}/torch/{name}")

    r = http.post(f"{PREFIX}/model/torch", json={
        "modelName": "ddp_cnn",
        "modulePath": "learningorchestra_amd.models.zoo",
        "class": "MnistCNN",
        "classParameters": {"channels": [32, 64], "fc_width": 256},
    })
    assert r.status_code == 201, r.text

    r = http.post(f"{PREFIX}/train/torch", json={
        "name": "ddp_fit",
        "modelName": "ddp_cnn",
        "method": "fit",
        "methodParameters": {
            "gpus": gpus,                       # <- the fan-out knob
            "x": "#numpy.random.rand(4096, 784).astype('float32')",
            "y": "#numpy.random.randint(0, 10, 4096)",
            "epochs": 2,
            "batch_size": 256,
        },
    })
    assert r.status_code == 201, r.text

    doc = http.get(f"{PREFIX}/observe/ddp_fit/wait",
                   params={"timeoutSeconds": 300}).json()["result"]
    print("finished:", doc.get("finished"), "exception:", doc.get("exception"))
    rows = http.get(f"{PREFIX}/train/torch/ddp_fit",
                    params={"limit": 10}).json()["result"]
    for d in rows:
        if d.get("_id", 0) >= 1:
            print("execution doc:", {k: d[k] for k in
                                     ("worldSize", "durationSeconds")
                                     if k in d})


if __name__ == "__main__":
    main()
