"""Minimal classification serving endpoint (beyond-reference deploy story:
the reference ships TRT/ncnn C++ inference demos, others/deploy — on ROCm the
native serving path is the framework itself behind an HTTP front).

Run:  python projects/others/deploy/serve.py --model resnet50 \
          [--weights ckpt.pth] [--num-classes 1000] [--port 8000]
Then: POST raw image bytes to /predict (curl --data-binary @img.jpg ...),
returns top-k class indices + softmax scores as JSON. GET /healthz for
probes, GET /metrics for prometheus. --max-batch N enables dynamic
micro-batching: concurrent requests are gathered for up to --batch-wait-ms
and run as ONE forward (the GPU-utilization lever for serving).

The app factory is importable (`create_app`) so tests drive it in-process
with fastapi.testclient — no socket needed.
"""
import argparse
import asyncio
import io

import torch
from fastapi import FastAPI, Request


class MicroBatcher:
    """Gather concurrent single-item requests into one batched forward.

    infer(x) enqueues a [1,...] tensor and awaits its slice of the batched
    output; the worker drains the queue up to max_batch items or
    max_wait_ms, whichever first. Started lazily on the running loop.
    """

    def __init__(self, forward_fn, max_batch=8, max_wait_ms=5.0):
        self.forward_fn = forward_fn
        self.max_batch = max_batch
        self.max_wait = max_wait_ms / 1000.0
        self.queue: asyncio.Queue = asyncio.Queue()
        self._task = None

    async def infer(self, x: torch.Tensor) -> torch.Tensor:
        if self._task is None or self._task.done():
            self._task = asyncio.get_running_loop().create_task(self._run())
        fut = asyncio.get_running_loop().create_future()
        await self.queue.put((x, fut))
        return await fut

    async def _run(self):
        loop = asyncio.get_running_loop()
        while True:
            x, fut = await self.queue.get()
            items = [(x, fut)]
            t0 = loop.time()
            while len(items) < self.max_batch:
                left = self.max_wait - (loop.time() - t0)
                if left <= 0:
                    break
                try:
                    items.append(await asyncio.wait_for(self.queue.get(),
                                                        timeout=left))
                except asyncio.TimeoutError:
                    break
            try:
                out = self.forward_fn(torch.cat([i[0] for i in items]))
                for (_, f), o in zip(items, out.split(1)):
                    if not f.done():
                        f.set_result(o)
            except Exception as e:  # propagate to every waiter
                for _, f in items:
                    if not f.done():
                        f.set_exception(e)


def create_app(model_name="resnet50", weights="", num_classes=1000,
               device=None, topk=5, image_size=224, task="cls",
               score_thresh=0.3, max_batch=1, batch_wait_ms=5.0):
    from deeplearning_amd.core.checkpoint import load_pretrained
    from deeplearning_amd.models import build_model

    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    device = torch.device(device)
    model = build_model(model_name, num_classes=num_classes).to(device)
    if weights:
        load_pretrained(model, weights)
    model.eval()
    if device.type == "cuda":
        model = model.to(memory_format=torch.channels_last)

    app = FastAPI(title="deeplearning_amd serve")
    app.state.model = model
    app.state.device = device

    # prometheus metrics (prometheus_client ships in the image); a private
    # registry so repeated create_app() calls (tests) don't collide
    from prometheus_client import (CONTENT_TYPE_LATEST, CollectorRegistry,
                                   Counter, Histogram, generate_latest)
    registry = CollectorRegistry()
    reqs = Counter("dla_serve_requests_total", "predict requests",
                   ["status"], registry=registry)
    lat = Histogram("dla_serve_latency_seconds", "predict latency",
                    registry=registry)

    def _preprocess(data: bytes) -> torch.Tensor:
        from PIL import Image

        from deeplearning_amd.data.transforms import pil_to_tensor
        img = Image.open(io.BytesIO(data)).convert("RGB")
        img = img.resize((image_size, image_size))
        x = pil_to_tensor(img).unsqueeze(0).to(device)
        if device.type == "cuda":
            x = x.contiguous(memory_format=torch.channels_last)
        return x

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "model": model_name,
                "device": str(device)}

    # raw body (python-multipart is not in the image, so no UploadFile):
    #   curl -X POST --data-binary @img.png http://host:8000/predict
    def _forward(x):
        with torch.no_grad():
            if device.type == "cuda":
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    return model([x[0]] if task == "det" else x)
            return model([x[0]] if task == "det" else x)

    batcher = MicroBatcher(_forward, max_batch, batch_wait_ms) \
        if (max_batch > 1 and task == "cls") else None

    @app.get("/metrics")
    def metrics():
        from fastapi import Response
        return Response(generate_latest(registry),
                        media_type=CONTENT_TYPE_LATEST)

    @app.post("/predict")
    async def predict(request: Request):
        import time
        t0 = time.perf_counter()
        data = await request.body()
        try:
            x = _preprocess(data)
            out = await batcher.infer(x) if batcher is not None \
                else _forward(x)
        except Exception:
            reqs.labels(status="error").inc()
            raise
        reqs.labels(status="ok").inc()
        lat.observe(time.perf_counter() - t0)
        if task == "det":
            det = out[0]
            keep = det["scores"] >= score_thresh
            return {"detections": [
                {"box": [round(v, 2) for v in b],
                 "class": int(l), "score": round(float(s), 4)}
                for b, s, l in zip(det["boxes"][keep].tolist(),
                                   det["scores"][keep].tolist(),
                                   det["labels"][keep].tolist())]}
        probs = out.float().softmax(-1)[0]
        k = min(topk, probs.numel())
        score, idx = probs.topk(k)
        return {"topk": [{"class": int(i), "score": float(s)}
                         for i, s in zip(idx.tolist(), score.tolist())]}

    return app


def main():
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--weights", default="")
    p.add_argument("--num-classes", type=int, default=1000)
    p.add_argument("--device", default=None)
    p.add_argument("--topk", type=int, default=5)
    p.add_argument("--img-size", type=int, default=224)
    p.add_argument("--task", default="cls", choices=["cls", "det"],
                   help="det: list-in/dict-out detectors (fasterrcnn_*, "
                        "retinanet_*, fcos_*); yolo models have their own "
                        "detect.py postprocess")
    p.add_argument("--score-thresh", type=float, default=0.3)
    p.add_argument("--max-batch", type=int, default=1,
                   help=">1 enables dynamic micro-batching (cls task)")
    p.add_argument("--batch-wait-ms", type=float, default=5.0)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args()
    app = create_app(args.model, args.weights, args.num_classes, args.device,
                     args.topk, args.img_size, args.task, args.score_thresh,
                     args.max_batch, args.batch_wait_ms)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
