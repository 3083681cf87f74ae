"""Inference server over a train-end export (TestClient, no sockets)."""

import tempfile

import pytest
import torch

try:
    from fastapi.testclient import TestClient
except ImportError:  # pragma: no cover
    TestClient = None

from elasticdl_amd.models import mnist
from elasticdl_amd.serving.server import ModelRunner, build_app


@pytest.mark.skipif(TestClient is None, reason="fastapi testclient missing")
def test_serve_exported_mnist_model():
    model = mnist.custom_model()
    with tempfile.NamedTemporaryFile(suffix=".pt") as f:
        torch.save(model.state_dict(), f.name)
        runner = ModelRunner("mnist", f.name, device="cpu")
        app = build_app(runner)
        client = TestClient(app)

        assert client.get("/health").json()["status"] == "ok"

        x, _ = mnist.synthetic_batch(2, seed=0)
        r = client.post("/v1/models/default:predict",
                        json={"instances": x.tolist()})
        assert r.status_code == 200, r.text
        preds = r.json()["predictions"]
        assert len(preds) == 2 and len(preds[0]) == 10
        # parity with direct eval
        with torch.no_grad():
            ref = model.eval()(x)
        assert torch.allclose(torch.tensor(preds), ref, atol=1e-5)

        bad = client.post("/v1/models/default:predict",
                          json={"instances": [["oops"]]})
        assert bad.status_code == 400

        # binary path: codec body in, codec body out, same numerics
        from elasticdl_amd.common import codec

        rb = client.post(
            "/v1/models/default:predict_binary",
            content=codec.encode({"instances": x}),
        )
        assert rb.status_code == 200, rb.text
        out = codec.decode(rb.content)["predictions"]
        assert torch.allclose(out, ref, atol=1e-5)
