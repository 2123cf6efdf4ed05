#!/usr/bin/env python3
"""Flagship benchmark: GLOM denoising training step (fwd + bwd + optimizer).

Metric (BASELINE.json): images/sec (whole node), Glom dim=512 levels=6
image_size=224 patch_size=14 iters=12, bf16, synthetic 224x224 images,
random-init weights. Weak scaling: fixed per-GPU batch (default 64).

Single GPU:   python bench.py --gpus 1 --steps K --warmup W
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=64, help="per-GPU batch")
    p.add_argument("--iters", type=int, default=12)
    p.add_argument("--dim", type=int, default=512)
    p.add_argument("--levels", type=int, default=6)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--patch-size", type=int, default=14)
    p.add_argument("--impl", choices=["native", "eager", "compile"],
                   default="native",
                   help="native = CDNA4 HIP engine; eager/compile = stock "
                        "PyTorch-ROCm running the same math (comparison)")
    p.add_argument("--micro", type=int, default=1,
                   help="pipelined microbatches per step (chunk i+1's "
                        "forward overlaps chunk i's backward; gradients "
                        "are exactly the full-batch gradient)")
    p.add_argument("--bucket-mb", type=int, default=16,
                   help="DP gradient all-reduce bucket size (MB); tune "
                        "against the 7-link xGMI ring bandwidth")
    p.add_argument("--mode", choices=["train", "infer"], default="train",
                   help="train = denoising fwd+bwd+AdamW (headline); "
                        "infer = no-grad forward under hipGraph replay")
    p.add_argument("--profile", action="store_true",
                   help="emit roctx ranges around each timed step (pair "
                        "with rocprofv3 --marker-trace)")
    return p.parse_args()


def main():
    args = parse_args()
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from glom_pytorch_amd import Glom
    from glom_pytorch_amd.parallel.trainer import DenoisingTrainer

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world > 1
    if distributed:
        from glom_pytorch_amd.parallel.failure import init_distributed
        init_distributed(timeout_s=300)
    torch.cuda.set_device(local_rank)
    dev = torch.device("cuda", local_rank)
    torch.manual_seed(1234 + rank)

    model = Glom(dim=args.dim, levels=args.levels, image_size=args.image_size,
                 patch_size=args.patch_size).to(dev, torch.bfloat16)
    if args.impl != "native":
        model.force_eager = True
    if args.impl == "compile":
        model = torch.compile(model)

    img = torch.randn(args.batch, 3, args.image_size, args.image_size,
                      device=dev, dtype=torch.bfloat16)

    def sync_all():
        if distributed:
            dist.barrier()
        torch.cuda.synchronize()

    if args.mode == "infer":
        if args.impl == "native":
            model.enable_graphs()
        loss = 0.0

        def step(x, iters):
            with torch.no_grad():
                model(x, iters=iters)
            return 0.0
    else:
        trainer = DenoisingTrainer(model, distributed=distributed,
                                   bucket_bytes=args.bucket_mb << 20,
                                   micro_batches=args.micro)

        def step(x, iters):
            # loss stays on-device inside the timed loop (no host sync);
            # converted once after the final synchronize
            return trainer.step(x, iters=iters, sync_loss=False)

    for _ in range(args.warmup):
        step(img, args.iters)

    sync_all()
    t0 = time.perf_counter()
    if args.profile:
        from glom_pytorch_amd.utils.profiling import trace_range
        for i in range(args.steps):
            with trace_range(f"bench/step{i}"):
                loss = step(img, args.iters)
    else:
        for _ in range(args.steps):
            loss = step(img, args.iters)
    sync_all()
    elapsed = time.perf_counter() - t0
    if torch.is_tensor(loss):
        loss = loss.item()

    # max over ranks
    if distributed:
        t = torch.tensor([elapsed], device=dev, dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    if rank == 0:
        n_gpus = world if distributed else 1
        global_batch = args.batch * n_gpus
        images_sec = global_batch * args.steps / elapsed
        print(json.dumps({
            "metric": "images/sec",
            "value": images_sec,
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": f"glom-{args.dim}x{args.levels}",
                "global_batch": global_batch,
                "image_size": args.image_size,
                "patch_size": args.patch_size,
                "iters": args.iters,
                "objective": ("denoising-mse fwd+bwd+adamw"
                              if args.mode == "train"
                              else "inference forward (hipGraph replay)"),
                # reference recipe (README.md:56-90): decode the top level
                # at t=7; iterations >= 7 run forward-only since their
                # gradient contribution is exactly zero (dead-graph
                # elimination, gradients bitwise-identical; applied to the
                # eager/compile comparison arms too)
                "decode_step": 7,
                "impl": args.impl,
                "parallelism": f"dp{n_gpus}",
            },
            "loss_last": loss,
        }))
    if distributed:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
