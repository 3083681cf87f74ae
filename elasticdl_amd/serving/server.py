"""Inference server for exported models.

The reference exports a SavedModel at train end for TF-Serving
(elasticdl/callbacks.py SavedModelExporter, docs/benchmark hybrid
training+serving clusters). The torch-native equivalent: load the
train-end export (state_dict) with its zoo module, serve
predictions over HTTP (FastAPI/uvicorn), bf16 on GPU.

    python -m elasticdl_amd.serving.server \
        --model_def mnist --model_path /ckpt/model.pt --port 8500

    POST /v1/models/default:predict   {"instances": [[...], ...]}
    GET  /health
"""

import argparse
from typing import Optional

import torch

from elasticdl_amd.common.args import parse_model_params
from elasticdl_amd.common.log_utils import default_logger as logger
from elasticdl_amd.utils.model_utils import get_model_spec


class ModelRunner:
    def __init__(self, model_def: str, model_path: str = "",
                 model_params: str = "", device: str = "auto",
                 dtype: Optional[torch.dtype] = None):
        if device == "auto":
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        spec = get_model_spec(model_def, parse_model_params(model_params))
        self.spec = spec
        self.model = spec.build_model()
        if model_path:
            state = torch.load(model_path, map_location="cpu",
                               weights_only=True)
            self.model.load_state_dict(state)
        if dtype is None:
            dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.dtype = dtype
        self.model = self.model.to(self.device, dtype).eval()
        self.input_dtype = self._infer_input_dtype()
        logger.info("Serving %s on %s (%s)", model_def, self.device, dtype)

    def _infer_input_dtype(self):
        mod = self.spec.module
        if hasattr(mod, "synthetic_batch"):
            x, _ = mod.synthetic_batch(1, seed=0)
            return x.dtype
        return torch.float32

    @torch.no_grad()
    def predict_tensor(self, x: torch.Tensor) -> torch.Tensor:
        if x.dtype.is_floating_point:
            x = x.to(self.device, self.dtype)
        else:
            x = x.to(self.device)
        return self.model(x).float().cpu()

    @torch.no_grad()
    def predict(self, instances) -> list:
        x = torch.as_tensor(instances, dtype=self.input_dtype)
        return self.predict_tensor(x).tolist()


def build_app(runner: ModelRunner):
    from fastapi import FastAPI, HTTPException, Request, Response
    from pydantic import BaseModel

    from elasticdl_amd.common import codec

    class PredictRequest(BaseModel):
        instances: list

    app = FastAPI(title="elasticdl_amd serving")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(runner.device)}

    @app.post("/v1/models/{name}:predict")
    def predict(name: str, req: PredictRequest):
        try:
            return {"predictions": runner.predict(req.instances)}
        except Exception as e:  # noqa: BLE001 - surface as 400
            raise HTTPException(status_code=400, detail=str(e))

    @app.post("/v1/models/{name}:predict_binary")
    async def predict_binary(name: str, request: Request):
        """Binary path: the request body is a codec-encoded
        {"instances": tensor}; the response body is codec-encoded
        {"predictions": tensor}. No per-element JSON marshaling — the
        framework codec's zero-copy frombuffer view feeds the model
        directly (10-100x faster for image-sized payloads)."""
        try:
            msg = codec.decode(await request.body())
            out = runner.predict_tensor(msg["instances"])
            return Response(
                content=codec.encode({"predictions": out}),
                media_type="application/octet-stream",
            )
        except Exception as e:  # noqa: BLE001
            raise HTTPException(status_code=400, detail=str(e))

    return app


def main(argv=None) -> int:
    ap = argparse.ArgumentParser("elasticdl serving")
    ap.add_argument("--model_def", required=True)
    ap.add_argument("--model_path", default="")
    ap.add_argument("--model_params", default="")
    ap.add_argument("--device", default="auto")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8500)
    args = ap.parse_args(argv)

    import uvicorn

    runner = ModelRunner(args.model_def, args.model_path, args.model_params,
                         args.device)
    uvicorn.run(build_app(runner), host=args.host, port=args.port)
    return 0


if __name__ == "__main__":
    main()
