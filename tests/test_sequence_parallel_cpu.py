"""Sequence parallelism (gloo, CPU, world_size 2): column-sharded forward
must equal the single-process eager forward, for both exchange modes
(all-gather and ring/online-softmax), including the -5e-4 self mask and the
local-radius mask applied at global column indices, stateful continuation,
and gradient flow through the differentiable all-gather."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from glom_pytorch_amd import Glom
from conftest import SMALL
from _netutil import free_port


def _worker(rank, world, port, cfg, img, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from glom_pytorch_amd.parallel.sequence import sp_forward, _shard_bounds
    try:
        _shard_bounds(17)
        raise SystemExit("expected ValueError for indivisible columns")
    except ValueError:
        pass
    torch.manual_seed(0)
    model = Glom(**cfg)
    res = {}
    with torch.no_grad():
        res["allgather"] = sp_forward(model, img, iters=3, mode="allgather",
                                      gather_output=True)
        res["ring"] = sp_forward(model, img, iters=3, mode="ring",
                                 gather_output=True)
        res["traj"] = sp_forward(model, img, iters=2, mode="ring",
                                 return_all=True, gather_output=True)
        # stateful continuation from a full-size levels tensor
        res["stateful"] = sp_forward(model, img, iters=2, mode="ring",
                                     levels=res["allgather"],
                                     gather_output=True)
    # SP training recipe: local-shard loss (ranks sum to the global mean),
    # backward through the differentiable all-gather, then all-reduce the
    # replicated weight grads exactly like DP.
    out = sp_forward(model, img, iters=2, mode="allgather")
    full_count = img.shape[0] * model.num_patches * model.dim
    loss = out[:, :, -1].pow(2).sum() / full_count
    loss.backward()
    for p in model.parameters():
        dist.all_reduce(p.grad)
    res["grads"] = {n: p.grad.clone() for n, p in model.named_parameters()}
    if rank == 0:
        torch.save(res, out_path)
    dist.destroy_process_group()


def _run_world(cfg, img, tmp_path, port):
    out_path = str(tmp_path / "sp.pt")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker,
                         args=(r, 2, port, cfg, img, out_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    return torch.load(out_path, weights_only=False)


@pytest.mark.timeout(300)
@pytest.mark.parametrize("radius", [0, 2])
def test_sp_forward_parity(tmp_path, radius):
    cfg = dict(SMALL)
    if radius:
        cfg["local_consensus_radius"] = radius
    torch.manual_seed(42)
    img = torch.randn(2, 3, 32, 32)
    got = _run_world(cfg, img, tmp_path, free_port())

    torch.manual_seed(0)
    model = Glom(**cfg)
    with torch.no_grad():
        ref3 = model(img, iters=3)
        traj = model(img, iters=2, return_all=True)
        stateful = model(img, iters=2, levels=ref3)
    assert torch.allclose(got["allgather"], ref3, atol=1e-5), \
        (got["allgather"] - ref3).abs().max()
    assert torch.allclose(got["ring"], ref3, atol=1e-5), \
        (got["ring"] - ref3).abs().max()
    assert torch.allclose(got["traj"], traj, atol=1e-5)
    assert torch.allclose(got["stateful"], stateful, atol=1e-5)

    # SP gradients: same loss over the full output => same weight grads
    out = model(img, iters=2)
    out[:, :, -1].pow(2).mean().backward()
    for n, p in model.named_parameters():
        assert torch.allclose(got["grads"][n], p.grad,
                              rtol=1e-4, atol=1e-6), n


def test_sp_ring_rejects_grad_mode():
    from glom_pytorch_amd.parallel.sequence import sp_forward
    model = Glom(**SMALL)
    img = torch.randn(1, 3, 32, 32, requires_grad=True)
    with pytest.raises(RuntimeError, match="inference-only"):
        sp_forward(model, img, iters=1, mode="ring")


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_sp_native_gpu_world1():
    """SP on a bf16 GPU rank runs the local FF work through the CDNA4
    kernels (GroupedFFFn/LevelMixFn); world-size-1 RCCL makes the gather
    trivial so the result must match the fused native forward."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(free_port())
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        from glom_pytorch_amd.parallel.sequence import sp_forward
        torch.manual_seed(0)
        model = Glom(**SMALL).to("cuda", torch.bfloat16)
        img = torch.randn(2, 3, 32, 32, device="cuda", dtype=torch.bfloat16)
        ref = model(img, iters=3)
        with torch.no_grad():
            ag = sp_forward(model, img, iters=3, mode="allgather",
                            gather_output=True)
            ring = sp_forward(model, img, iters=3, mode="ring",
                              gather_output=True)

        def rel(a, b):
            return ((a.float() - b.float()).norm()
                    / b.float().norm().clamp_min(1e-8)).item()
        assert rel(ag, ref) < 1.5e-2, rel(ag, ref)
        assert rel(ring, ref) < 1.5e-2, rel(ring, ref)

        # grads flow through the SP training path on GPU too
        out = sp_forward(model, img, iters=2, mode="allgather")
        out[:, :, -1].float().pow(2).mean().backward()
        g = model.bottom_up.net[1].weight.grad
        assert g is not None and torch.isfinite(g.float()).all()
    finally:
        dist.destroy_process_group()
