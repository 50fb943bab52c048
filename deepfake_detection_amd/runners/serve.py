"""Inference serving: a FastAPI endpoint around the fp16 deepfake model.

The reference ships no serving layer ("no serving layer");
this is MI355X-stack value-add for production deployment. Single-process,
one GPU, micro-batched: concurrent requests queue and run as one forward
per batch window (288 GB HBM3E holds the whole model + large batches, so
batching is bounded by latency, not memory).

Run:  python -m deepfake_detection_amd.runners.serve --checkpoint model_half.pth.tar
Then: curl -F "file=@img.png" localhost:8000/predict
"""

import argparse
import asyncio
import io
import logging
import time

import numpy as np
import torch

from .. import params
from ..models import create_deepfake_model_v4

_logger = logging.getLogger(__name__)


class InferenceEngine:
    """Owns the model and runs micro-batched forwards."""

    def __init__(self, checkpoint="", model_name="efficientnet_deepfake_v4",
                 device=None, fp16=True, max_batch=32, batch_window_ms=3.0):
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        self.dtype = torch.float16 if (fp16 and self.device == "cuda") else torch.float32
        model = create_deepfake_model_v4(
            model_name, num_classes=2, in_chans=12,
            checkpoint_path=checkpoint, strict=False)
        model = params.DeepFakeModel(model).to(self.device).to(self.dtype).eval()
        if self.device == "cuda":
            model = model.to(memory_format=torch.channels_last)
        self.model = model
        self.max_batch = max_batch
        self.batch_window_s = batch_window_ms / 1000.0
        self._queue: "asyncio.Queue" = asyncio.Queue()
        self._worker = None

    def preprocess(self, image_bytes: bytes) -> torch.Tensor:
        from PIL import Image

        img = Image.open(io.BytesIO(image_bytes)).convert("RGB")
        arr = np.asarray(img)
        return params.preprocess_image(arr, device="cpu", dtype=torch.float32)

    @torch.no_grad()
    def forward(self, batch: torch.Tensor) -> torch.Tensor:
        x = batch.to(self.device, dtype=self.dtype)
        if self.device == "cuda":
            x = x.contiguous(memory_format=torch.channels_last)
        scores = self.model(x)
        return scores[:, 0].float().cpu()  # fake probability (class 0 = fake)

    async def submit(self, tensor: torch.Tensor) -> float:
        fut = asyncio.get_running_loop().create_future()
        await self._queue.put((tensor, fut))
        return await fut

    async def _batch_loop(self):
        while True:
            tensor, fut = await self._queue.get()
            items = [(tensor, fut)]
            deadline = time.monotonic() + self.batch_window_s
            while len(items) < self.max_batch:
                timeout = deadline - time.monotonic()
                if timeout <= 0:
                    break
                try:
                    items.append(await asyncio.wait_for(self._queue.get(), timeout))
                except asyncio.TimeoutError:
                    break
            batch = torch.cat([t for t, _ in items], dim=0)
            try:
                scores = await asyncio.get_running_loop().run_in_executor(
                    None, self.forward, batch)
                for (_, f), s in zip(items, scores.tolist()):
                    if not f.done():
                        f.set_result(s)
            except Exception as e:  # noqa: BLE001
                for _, f in items:
                    if not f.done():
                        f.set_exception(e)

    def start(self):
        self._worker = asyncio.get_running_loop().create_task(self._batch_loop())


def create_app(engine: InferenceEngine):
    from fastapi import FastAPI, Request

    app = FastAPI(title="deepfake_detection_amd", version="0.1.0")

    @app.on_event("startup")
    async def _startup():
        engine.start()

    @app.get("/health")
    async def health():
        return {"status": "ok", "device": engine.device,
                "dtype": str(engine.dtype).replace("torch.", "")}

    # raw image bytes in the request body (PNG/JPEG): no multipart dependency
    #   curl --data-binary @img.png -H "Content-Type: application/octet-stream" \
    #        localhost:8000/predict
    @app.post("/predict")
    async def predict(request: Request):
        data = await request.body()
        tensor = engine.preprocess(data)
        score = await engine.submit(tensor)
        return {"fake_probability": score, "label": "fake" if score >= 0.5 else "real"}

    return app


def main(argv=None):
    p = argparse.ArgumentParser(description="deepfake inference server")
    p.add_argument("--checkpoint", default="", help="model_half.pth.tar path")
    p.add_argument("--model", default="efficientnet_deepfake_v4")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--max-batch", type=int, default=32)
    p.add_argument("--batch-window-ms", type=float, default=3.0)
    p.add_argument("--no-fp16", action="store_true")
    args = p.parse_args(argv)

    import uvicorn

    engine = InferenceEngine(
        checkpoint=args.checkpoint, model_name=args.model, fp16=not args.no_fp16,
        max_batch=args.max_batch, batch_window_ms=args.batch_window_ms)
    uvicorn.run(create_app(engine), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
