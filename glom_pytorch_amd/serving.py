"""Minimal production serving for the GLOM engine.

A FastAPI app around a single-GPU model with dynamic micro-batching:
concurrent requests are coalesced (up to ``max_batch`` or ``max_wait_ms``)
into one forward, which runs on the CDNA4 engine under hipGraph replay
(fixed batch shapes are padded so every batch replays a captured graph).

    from glom_pytorch_amd.serving import create_app
    app = create_app(model, iters=12)        # uvicorn some_module:app

POST /embed with a raw little-endian f32 body of shape (3, H, W) returns
the top-level embedding (n_patches x dim) as f32 bytes; /healthz reports
liveness and queue depth.
"""

import asyncio
import concurrent.futures

import numpy as np
import torch


class MicroBatcher:
    """Coalesce concurrent single-image requests into one padded forward."""

    def __init__(self, model, iters: int, max_batch: int = 16,
                 max_wait_ms: float = 2.0):
        self.model = model
        self.iters = iters
        self.max_batch = max_batch
        self.max_wait = max_wait_ms / 1e3
        self.queue: asyncio.Queue = asyncio.Queue()
        self._task = None
        # single worker thread: keeps the event loop responsive during the
        # GPU forward + device sync while preserving batch ordering
        self._pool = concurrent.futures.ThreadPoolExecutor(max_workers=1)
        p = next(model.parameters())
        self.device, self.dtype = p.device, p.dtype
        # fixed padded batch shapes -> every batch replays a captured graph
        if (p.is_cuda and hasattr(model, "enable_graphs")
                and getattr(model, "_graph_cache", None) is None):
            model.enable_graphs()

    def start(self):
        self._task = asyncio.get_event_loop().create_task(self._loop())

    async def stop(self):
        if self._task:
            self._task.cancel()
        self._pool.shutdown(wait=False)

    async def submit(self, img: torch.Tensor) -> torch.Tensor:
        fut = asyncio.get_event_loop().create_future()
        await self.queue.put((img, fut))
        return await fut

    async def _loop(self):
        while True:
            img, fut = await self.queue.get()
            batch = [(img, fut)]
            deadline = asyncio.get_event_loop().time() + self.max_wait
            while len(batch) < self.max_batch:
                timeout = deadline - asyncio.get_event_loop().time()
                if timeout <= 0:
                    break
                try:
                    batch.append(await asyncio.wait_for(self.queue.get(),
                                                        timeout))
                except asyncio.TimeoutError:
                    break
            try:
                outs = await asyncio.get_event_loop().run_in_executor(
                    self._pool, self._run, [b[0] for b in batch])
                for (_, f), out in zip(batch, outs):
                    if not f.done():
                        f.set_result(out)
            except Exception as e:   # pragma: no cover - error propagation
                for _, f in batch:
                    if not f.done():
                        f.set_exception(e)

    def _run(self, imgs):
        n = len(imgs)
        # pad to the next power of two so hipGraph capture is reused across
        # a handful of batch shapes instead of one graph per batch size
        padded = 1
        while padded < n:
            padded *= 2
        x = torch.stack(imgs + [imgs[-1]] * (padded - n))
        x = x.to(self.device, self.dtype, non_blocking=True)
        with torch.no_grad():
            levels = self.model(x, iters=self.iters)
        top = levels[:n, :, -1].float().cpu()   # (n, n_patches, dim)
        return list(top)


def create_app(model, iters: int = 12, max_batch: int = 16,
               max_wait_ms: float = 2.0):
    from contextlib import asynccontextmanager

    from fastapi import FastAPI, Request, Response

    batcher = MicroBatcher(model, iters, max_batch, max_wait_ms)
    size = model.image_size

    @asynccontextmanager
    async def _lifespan(app):
        batcher.start()
        yield
        await batcher.stop()

    app = FastAPI(title="glom_pytorch_amd", lifespan=_lifespan)

    @app.get("/healthz")
    async def healthz():
        return {"ok": True, "queued": batcher.queue.qsize(),
                "image_size": size, "iters": iters}

    @app.post("/embed")
    async def embed(request: Request):
        body = await request.body()
        expected = 3 * size * size * 4
        if len(body) != expected:
            return Response(
                status_code=400,
                content=(f"expected {expected} bytes "
                         f"(f32 3x{size}x{size}), got {len(body)}"))
        img = np.frombuffer(body, dtype=np.float32).reshape(3, size, size)
        out = await batcher.submit(torch.from_numpy(img.copy()))
        return Response(content=out.numpy().tobytes(),
                        media_type="application/octet-stream",
                        headers={"x-shape": f"{out.shape[0]},{out.shape[1]}"})

    return app
