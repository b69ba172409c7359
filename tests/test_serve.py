"""Inference HTTP service: the full pipeline behind FastAPI (CPU, tiny net)."""
import io

import numpy as np
import pytest
import torch

fastapi = pytest.importorskip("fastapi")

from improved_body_parts_amd.serve import PoseService, create_app, decode_image


@pytest.fixture(scope="module")
def client():
    from fastapi.testclient import TestClient
    svc = PoseService(config_name="Canonical", nstack=1, device="cpu",
                      bf16=False)
    # shrink the working resolution so the CPU eager path stays fast
    svc.model_params = dict(svc.model_params)
    svc.model_params["boxsize"] = 128
    return TestClient(create_app(service=svc))


def test_healthz_and_info(client):
    h = client.get("/healthz").json()
    assert h["status"] == "ok"
    i = client.get("/info").json()
    assert i["num_parts"] == 18 and i["num_limbs"] == 30


def test_pose_npy_roundtrip(client):
    img = np.random.RandomState(0).rand(128, 128, 3).astype(np.float32)
    buf = io.BytesIO()
    np.save(buf, img)
    r = client.post("/pose", content=buf.getvalue())
    assert r.status_code == 200, r.text
    j = r.json()
    assert j["image_size"] == [128, 128]
    assert isinstance(j["people"], list)
    for person in j["people"]:
        assert "score" in person and len(person["keypoints"]) >= 17


def test_pose_rejects_garbage(client):
    r = client.post("/pose", content=b"not an image at all")
    assert r.status_code == 400


def test_decode_image_json_and_uint8():
    img = (np.random.RandomState(1).rand(8, 8, 3) * 255).astype(np.uint8)
    import json
    arr = decode_image(json.dumps({"image": img.tolist()}).encode(),
                       "application/json")
    assert arr.shape == (8, 8, 3) and arr.max() <= 1.0


def test_decode_image_png():
    from PIL import Image
    img = Image.fromarray((np.random.RandomState(2).rand(16, 16, 3) * 255)
                          .astype(np.uint8))
    buf = io.BytesIO()
    img.save(buf, format="PNG")
    arr = decode_image(buf.getvalue())
    assert arr.shape == (16, 16, 3) and 0.0 <= arr.min() and arr.max() <= 1.0
