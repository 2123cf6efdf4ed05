"""Multi-process (gloo, CPU) tests for the data-parallel layer: 2-rank
bucketed all-reduce gradients must equal single-process big-batch gradients
(SURVEY.md §4 item 5)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from glom_pytorch_amd import Glom
from conftest import SMALL
from _netutil import free_port


def _grads_single(img_all):
    torch.manual_seed(0)
    model = Glom(**SMALL)
    out = model(img_all, iters=2, return_all=True)
    loss = out[2, :, :, -1].pow(2).mean()
    loss.backward()
    return {n: p.grad.clone() for n, p in model.named_parameters()}


def _worker(rank, world, port, img_all, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    from glom_pytorch_amd.parallel.ddp import BucketedDDP
    model = Glom(**SMALL)
    ddp = BucketedDDP(model, bucket_bytes=1 << 20)
    shard = img_all.chunk(world)[rank]
    out = model(shard, iters=2, return_all=True)
    loss = out[2, :, :, -1].pow(2).mean()
    loss.backward()
    ddp.finalize()
    if rank == 0:
        torch.save({n: p.grad.clone() for n, p in model.named_parameters()},
                   out_path)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_grad_parity_vs_single_process(tmp_path):
    torch.manual_seed(42)
    img_all = torch.randn(4, 3, 32, 32)
    ref = _grads_single(img_all)

    out_path = str(tmp_path / "grads.pt")
    ctx = mp.get_context("spawn")
    port = free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, img_all, out_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    got = torch.load(out_path, weights_only=False)

    # mean over 2 equal shards of the per-shard mean losses == big-batch mean
    for n, g in ref.items():
        assert torch.allclose(got[n], g, rtol=1e-4, atol=1e-6), n


def test_trainer_step_cpu():
    from glom_pytorch_amd.parallel.trainer import DenoisingTrainer
    torch.manual_seed(0)
    model = Glom(**SMALL)
    tr = DenoisingTrainer(model, noise_std=0.5, decode_step=2)
    img = torch.randn(2, 3, 32, 32)
    l1 = tr.step(img, iters=3)
    l2 = tr.step(img, iters=3)
    assert l1 > 0 and l2 > 0 and tr.step_idx == 2


def test_trainer_checkpoint_roundtrip(tmp_path):
    from glom_pytorch_amd.parallel.trainer import DenoisingTrainer
    torch.manual_seed(0)
    model = Glom(**SMALL)
    tr = DenoisingTrainer(model, noise_std=0.5, decode_step=2)
    img = torch.randn(2, 3, 32, 32)
    tr.step(img, iters=3)
    path = str(tmp_path / "ckpt.pt")
    tr.save_checkpoint(path)

    model2 = Glom(**SMALL)
    tr2 = DenoisingTrainer(model2, noise_std=0.5, decode_step=2)
    tr2.load_checkpoint(path)
    assert tr2.step_idx == 1
    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  model2.named_parameters()):
        assert n1 == n2 and torch.equal(p1, p2)
    # identical continuation from identical RNG + weights
    la = tr.step(img, iters=3)
    lb = tr2.step(img, iters=3)
    assert abs(la - lb) < 1e-6


def test_trainer_deferred_loss_sync():
    """sync_loss=False returns the loss as a detached tensor (no host
    sync) — the bench path; values must agree with the synced path."""
    from glom_pytorch_amd.parallel.trainer import DenoisingTrainer
    torch.manual_seed(0)
    model = Glom(**SMALL)
    tr = DenoisingTrainer(model, noise_std=0.5, decode_step=2)
    img = torch.randn(2, 3, 32, 32)
    t = tr.step(img, iters=2, sync_loss=False)
    assert torch.is_tensor(t) and not t.requires_grad
    assert t.item() > 0


def _worker_partial(rank, world, port, img_all, out_path):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    from glom_pytorch_amd.parallel.ddp import BucketedDDP
    model = Glom(**SMALL)
    ddp = BucketedDDP(model, bucket_bytes=1 << 30)   # ONE bucket
    shard = img_all.chunk(world)[rank]
    # stateful path: init_levels gets NO grad, so the single bucket is
    # partially filled and must be flushed by finalize()
    with torch.no_grad():
        warm = model(shard, iters=1)
    out = model(shard, iters=2, levels=warm)
    out[:, :, -1].pow(2).mean().backward()
    assert model.init_levels.grad is None
    ddp.finalize()
    if rank == 0:
        torch.save({n: (p.grad.clone() if p.grad is not None else None)
                    for n, p in model.named_parameters()}, out_path)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_partial_bucket_stateful_path(tmp_path):
    """A bucket with a no-grad param (init_levels under levels=) must
    still all-reduce the params that DO have grads."""
    torch.manual_seed(42)
    img_all = torch.randn(4, 3, 32, 32)
    out_path = str(tmp_path / "pg.pt")
    _pp = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker_partial,
                         args=(r, 2, _pp, img_all, out_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    got = torch.load(out_path, weights_only=False)

    # single-process reference over the full batch
    torch.manual_seed(0)
    model = Glom(**SMALL)
    with torch.no_grad():
        warm = model(img_all, iters=1)
    out = model(img_all, iters=2, levels=warm)
    out[:, :, -1].pow(2).mean().backward()
    for n, p in model.named_parameters():
        if n == "init_levels":
            # flushed as zeros on ranks; single-process has None
            assert got[n] is None or torch.count_nonzero(got[n]) == 0
            continue
        assert torch.allclose(got[n], p.grad, rtol=1e-4, atol=1e-6), n
