"""Pose-estimation inference service (FastAPI).

The reference ships only scripts; this framework is built for production
serving on MI355X, so the full device-resident pipeline (predict -> HIP
peak/limb kernels -> greedy assembly) is exposed behind an HTTP API:

    POST /pose    image (PNG/JPEG bytes, .npy array, or JSON) -> people
    GET  /healthz liveness + device/native-path report
    GET  /info    model/config summary

Run: ``python scripts/serve.py --checkpoint ckpt.pth [--port 8000]``
(random-init weights without a checkpoint — useful for plumbing tests).

A single model instance is shared; requests serialise on a lock (the GPU
pipeline is throughput-bound, not latency-bound — batching across requests
is a deliberate non-goal at this model size: one 512^2 forward is ~3 ms).
"""
import io
import threading
from typing import Optional

import numpy as np
import torch

try:  # the service degrades gracefully when fastapi is absent
    from fastapi import FastAPI, Request, Response
except Exception:  # pragma: no cover
    FastAPI = Request = Response = None

from .config import GetConfig, TrainingOpt
from .config.inference_params import InferenceParams
from .engine.inference import process
from .models import NetworkEval


class PoseService:
    """Owns the model + config and turns raw images into keypoint lists."""

    def __init__(self, config_name: str = "Canonical", nstack: int = 4,
                 checkpoint: Optional[str] = None, device: Optional[str] = None,
                 bf16: Optional[bool] = None):
        self.config = GetConfig(config_name)
        self.opt = TrainingOpt(nstack=nstack, batch_size=1)
        self.device = torch.device(device or
                                   ("cuda" if torch.cuda.is_available() else "cpu"))
        use_bf16 = bf16 if bf16 is not None else self.device.type == "cuda"
        model = NetworkEval(self.opt, self.config, bn=True)
        if checkpoint:
            state = torch.load(checkpoint, map_location="cpu",
                               weights_only=False)
            weights = state.get("weights", state)
            weights = {k.replace("module.", "", 1): v for k, v in weights.items()}
            model.posenet.load_state_dict(weights)
        model = model.to(self.device)
        if use_bf16:
            model = model.bfloat16()
            for m in model.modules():
                if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                    m.float()
        model.eval()
        self.model = model
        self.params, self.model_params = InferenceParams().as_params_dict()
        self._lock = threading.Lock()

    def infer(self, image: np.ndarray):
        """image (H, W, 3) float32 in [0,1] -> list of people dicts."""
        with self._lock, torch.no_grad():
            people = process(image, self.model, self.config, self.params,
                             self.model_params)
        out = []
        for kps, score in people:
            out.append({
                "score": float(score),
                "keypoints": [None if pt is None else
                              [float(pt[0]), float(pt[1])] for pt in kps],
            })
        return out


def decode_image(data: bytes, content_type: str = "") -> np.ndarray:
    """Accept PNG/JPEG bytes, a .npy array, or a JSON nested list."""
    if data[:6] == b"\x93NUMPY":
        arr = np.load(io.BytesIO(data), allow_pickle=False)
    elif content_type.startswith("application/json") or data[:1] in (b"[", b"{"):
        import json
        obj = json.loads(data)
        arr = np.asarray(obj["image"] if isinstance(obj, dict) else obj,
                         dtype=np.float32)
    else:
        from PIL import Image
        arr = np.asarray(Image.open(io.BytesIO(data)).convert("RGB"),
                         dtype=np.float32) / 255.0
    arr = np.asarray(arr, dtype=np.float32)
    if arr.ndim != 3 or arr.shape[2] != 3:
        raise ValueError(f"expected (H, W, 3) image, got {arr.shape}")
    if arr.max() > 1.5:  # 0..255 input
        arr = arr / 255.0
    return np.ascontiguousarray(arr)


def create_app(service: Optional[PoseService] = None, **service_kwargs):
    if FastAPI is None:  # pragma: no cover
        raise RuntimeError("fastapi is required for the HTTP service")
    app = FastAPI(title="improved-body-parts-amd pose service")
    svc = service or PoseService(**service_kwargs)
    app.state.service = svc

    @app.get("/healthz")
    def healthz():
        from .ops import _backend
        return {
            "status": "ok",
            "device": str(svc.device),
            "native_hip": bool(svc.device.type == "cuda"
                               and _backend.hip_available()),
        }

    @app.get("/info")
    def info():
        return {
            "model": f"{svc.opt.nstack}-stage IMHN",
            "input": [svc.config.height, svc.config.width],
            "num_parts": svc.config.num_parts,
            "num_limbs": len(svc.config.limbs_conn),
        }

    @app.post("/pose")
    async def pose(request: Request, response: Response):
        data = await request.body()
        try:
            img = decode_image(data, request.headers.get("content-type", ""))
        except Exception as e:
            response.status_code = 400
            return {"error": str(e)}
        people = svc.infer(img)
        return {"people": people, "image_size": list(img.shape[:2])}

    return app
