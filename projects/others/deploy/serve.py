"""Minimal classification serving endpoint (beyond-reference deploy story:
the reference ships TRT/ncnn C++ inference demos, others/deploy — on ROCm the
native serving path is the framework itself behind an HTTP front).

Run:  python projects/others/deploy/serve.py --model resnet50 \
          [--weights ckpt.pth] [--num-classes 1000] [--port 8000]
Then: POST an image file to /predict (multipart field "file"), returns
      top-k class indices + softmax scores as JSON. GET /healthz for probes.

The app factory is importable (`create_app`) so tests drive it in-process
with fastapi.testclient — no socket needed.
"""
import argparse
import io

import torch
from fastapi import FastAPI, Request


def create_app(model_name="resnet50", weights="", num_classes=1000,
               device=None, topk=5, image_size=224, task="cls",
               score_thresh=0.3):
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
            out = _forward(x)
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
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args()
    app = create_app(args.model, args.weights, args.num_classes, args.device,
                     args.topk, args.img_size, args.task, args.score_thresh)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
