"""Serving layer: micro-batched embed endpoint (CPU eager model here; the
same app serves the CDNA4 engine under hipGraph replay on GPU)."""

import numpy as np
import pytest
import torch

from glom_pytorch_amd import Glom
from conftest import SMALL

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from glom_pytorch_amd.serving import create_app  # noqa: E402


def test_embed_endpoint_matches_direct_forward():
    torch.manual_seed(0)
    model = Glom(**SMALL)
    app = create_app(model, iters=3, max_wait_ms=1.0)
    with TestClient(app) as client:
        r = client.get("/healthz")
        assert r.status_code == 200 and r.json()["ok"]

        img = np.random.RandomState(0).randn(3, 32, 32).astype(np.float32)
        r = client.post("/embed", content=img.tobytes())
        assert r.status_code == 200
        n, d = map(int, r.headers["x-shape"].split(","))
        out = np.frombuffer(r.content, dtype=np.float32).reshape(n, d)

        with torch.no_grad():
            ref = model(torch.from_numpy(img)[None], iters=3)[0, :, -1]
        assert np.allclose(out, ref.numpy(), atol=1e-5)


def test_embed_microbatching_concurrent():
    import concurrent.futures
    torch.manual_seed(0)
    model = Glom(**SMALL)
    app = create_app(model, iters=2, max_batch=4, max_wait_ms=20.0)
    imgs = [np.random.RandomState(i).randn(3, 32, 32).astype(np.float32)
            for i in range(6)]
    with TestClient(app) as client:
        def post(i):
            return client.post("/embed", content=imgs[i].tobytes())
        with concurrent.futures.ThreadPoolExecutor(6) as ex:
            rs = list(ex.map(post, range(6)))
        for i, r in enumerate(rs):
            assert r.status_code == 200
            n, d = map(int, r.headers["x-shape"].split(","))
            out = np.frombuffer(r.content, dtype=np.float32).reshape(n, d)
            with torch.no_grad():
                ref = model(torch.from_numpy(imgs[i])[None], iters=2)[0, :, -1]
            assert np.allclose(out, ref.numpy(), atol=1e-5), i


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_embed_endpoint_gpu_graphs():
    torch.manual_seed(0)
    model = Glom(**SMALL).to("cuda", torch.bfloat16).enable_graphs()
    app = create_app(model, iters=3, max_wait_ms=1.0)
    with TestClient(app) as client:
        img = np.random.RandomState(1).randn(3, 32, 32).astype(np.float32)
        r = client.post("/embed", content=img.tobytes())
        assert r.status_code == 200
        n, d = map(int, r.headers["x-shape"].split(","))
        out = np.frombuffer(r.content, dtype=np.float32).reshape(n, d)
        with torch.no_grad():
            ref = model(torch.from_numpy(img)[None].to("cuda",
                        torch.bfloat16), iters=3)[0, :, -1]
        assert np.allclose(out, ref.float().cpu().numpy(), atol=1e-2)


def test_embed_rejects_malformed_body():
    model = Glom(**SMALL)
    app = create_app(model, iters=2)
    with TestClient(app) as client:
        r = client.post("/embed", content=b"\x00" * 100)
        assert r.status_code == 400
        assert b"expected" in r.content


def test_healthz_live_during_slow_inference():
    """The forward runs in a worker thread (run_in_executor), so /healthz
    answers while a request is being served — the liveness probe measures
    liveness, not inference latency (round-1 ADVICE item)."""
    import threading
    import time

    torch.manual_seed(0)
    model = Glom(**SMALL)
    size = model.image_size

    slow_gate = threading.Event()

    class SlowGlom(torch.nn.Module):
        def __init__(self, inner):
            super().__init__()
            self.inner = inner
            self.image_size = inner.image_size

        def parameters(self, *a, **k):
            return self.inner.parameters(*a, **k)

        def forward(self, *a, **k):
            slow_gate.wait(timeout=5.0)   # block the WORKER thread only
            return self.inner(*a, **k)

    app = create_app(SlowGlom(model), iters=2, max_wait_ms=0.5)
    with TestClient(app) as client:
        body = np.zeros((3, size, size), dtype=np.float32).tobytes()
        results = {}

        def post():
            results["embed"] = client.post("/embed", content=body)

        th = threading.Thread(target=post)
        th.start()
        time.sleep(0.3)    # request is now blocked inside the worker
        t0 = time.perf_counter()
        h = client.get("/healthz")
        dt = time.perf_counter() - t0
        assert h.status_code == 200 and h.json()["ok"]
        assert dt < 1.0, f"healthz took {dt:.2f}s while inference ran"
        slow_gate.set()
        th.join(timeout=10)
        assert results["embed"].status_code == 200


def test_microbatcher_pads_to_power_of_two():
    """Batch shapes are padded to the next power of two so hipGraph
    capture is reused across a handful of shapes instead of one graph
    per batch size."""
    torch.manual_seed(0)
    model = Glom(**SMALL)
    from glom_pytorch_amd.serving import MicroBatcher
    mb = MicroBatcher(model, iters=2)
    seen = []
    orig_forward = model.forward

    def spy(img, *a, **k):
        seen.append(img.shape[0])
        return orig_forward(img, *a, **k)

    model.forward = spy
    size = model.image_size
    imgs = [torch.randn(3, size, size) for _ in range(3)]
    outs = mb._run(imgs)
    assert seen == [4]            # 3 requests -> padded to 4
    assert len(outs) == 3         # padding rows dropped from results
    assert outs[0].shape == (model.num_patches, model.dim)
