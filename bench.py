#!/usr/bin/env python3
"""Benchmark: the BASELINE.json headline metric — images/sec (whole node),
MobileNetV3-Large 1.0x 224 bf16 training, synthetic data, random-init
weights, N GPUs of one node (DP over RCCL/xGMI).

Single GPU:   python bench.py --steps 30 --warmup 10
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-job aggregate.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from yet_another_mobilenet_series_amd import models  # noqa: E402
from yet_another_mobilenet_series_amd.ops import functional as OF  # noqa: E402
from yet_another_mobilenet_series_amd.ops.backend import (  # noqa: E402
    load_extension, set_engine_mode)
from yet_another_mobilenet_series_amd.parallel import (  # noqa: E402
    DataParallelEngine, distributed as udist)
from yet_another_mobilenet_series_amd.utils.config import default_config  # noqa: E402
from yet_another_mobilenet_series_amd.utils.optim import (  # noqa: E402
    ExponentialMovingAverage, LRScheduler, get_optimizer)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="mobilenet_v3_large")
    p.add_argument("--width", type=float, default=1.0)
    p.add_argument("--batch", type=int, default=256,
                   help="per-GPU batch (weak scaling)")
    p.add_argument("--engine", default="hip", choices=["hip", "reference"],
                   help="reference = eager fp32->bf16 torch ops (A/B baseline)")
    p.add_argument("--graph", action="store_true", default=False,
                   help="capture the step in a hipGraph (torch.cuda.CUDAGraph)")
    # defaults match the flagship recipe (apps/mobilenet_v3_large.yml):
    # RMSpropTF + EMA 0.9999 + per-step LR schedule all INSIDE the timed
    # step (VERDICT round-1 weak #4: the reference's hot loop runs them
    # every step, so the metric must include them)
    p.add_argument("--optimizer", default="rmsprop",
                   choices=["sgd", "rmsprop"])
    p.add_argument("--ema", dest="ema", action="store_true", default=True)
    p.add_argument("--no-ema", dest="ema", action="store_false")
    p.add_argument("--bucket-mb", type=float, default=None,
                   help="per-bucket MB for the gradient all-reduce "
                        "(default: auto from model size, see ddp.py)")
    return p.parse_args()


def main():
    args = parse_args()
    rank = udist.init_dist("nccl" if torch.cuda.is_available() else "gloo")
    world = udist.get_world_size()
    dev = (torch.device("cuda", udist.get_local_rank())
           if torch.cuda.is_available() else torch.device("cpu"))
    on_gpu = dev.type == "cuda"
    if on_gpu:
        torch.cuda.set_device(dev)
        load_extension(required=args.engine == "hip")
    set_engine_mode(args.engine)

    torch.manual_seed(1234 + rank)
    model = models.build_model(args.model, dropout=0.0,
                               width_mult=args.width).to(dev)
    if on_gpu:
        model = model.to(memory_format=torch.channels_last)
    engine = DataParallelEngine(model, bucket_mb=args.bucket_mb, overlap=True)
    ocfg = default_config().optim
    ocfg.name = args.optimizer
    if args.optimizer == "sgd":
        ocfg.lr, ocfg.weight_decay = 0.4, 4e-5
        ocfg.lr_schedule = "cosine"
    else:  # the V3 paper recipe (apps/mobilenet_v3_large.yml)
        ocfg.lr, ocfg.weight_decay = 0.064, 1e-5
        ocfg.lr_schedule = "exp"
        ocfg.exp_decay_rate, ocfg.exp_decay_epochs = 0.973, 2.4
    opt = get_optimizer(model, ocfg)  # fused multi-tensor kernels on GPU
    # recipe LR schedule (warmup + exp decay), advanced once per step —
    # host-side arithmetic, but part of the reference's hot loop
    ocfg.warmup_epochs = 3
    lr_sched = LRScheduler(opt, ocfg, steps_per_epoch=5004, epochs=350)
    ema = (ExponentialMovingAverage(model, decay=0.9999)
           if args.ema else None)
    model.train()

    x = torch.randn(args.batch, 3, 224, 224, device=dev)
    if on_gpu:
        x = x.bfloat16().contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (args.batch,), device=dev)

    import contextlib

    def _fwd_ctx():
        # reference engine on GPU: eager torch ops under bf16 autocast
        # (the honest MIOpen-based baseline the HIP engine must beat)
        if args.engine == "reference" and on_gpu:
            return torch.autocast("cuda", torch.bfloat16)
        return contextlib.nullcontext()

    step_no = [0]

    def step():
        lr_sched.step(step_no[0])
        step_no[0] += 1
        engine.zero_grad()
        with _fwd_ctx():
            logits = engine(x)
        loss = OF.label_smooth_ce(logits.float() if args.engine == "reference"
                                  else logits, y, 0.1)
        loss.backward()
        engine.finish()
        opt.step()
        if ema is not None:
            ema.update()
        return loss.detach()

    graphed = None
    if args.graph and on_gpu:
        # warm up allocator + RCCL before capture
        for _ in range(3):
            step()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            static_loss = step()
        graphed = (g, static_loss)

    def run_one():
        if graphed is not None:
            graphed[0].replay()
            return graphed[1]
        return step()

    for _ in range(max(args.warmup, 1)):
        run_one()

    udist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    loss = None
    for _ in range(args.steps):
        loss = run_one()
    udist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    # max over ranks = whole-job wall time
    t = torch.tensor([dt], device=dev if on_gpu else "cpu")
    if world > 1:
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    dt = t.item()
    ms_per_step = dt / args.steps * 1000.0
    imgs_per_sec = args.batch * world * args.steps / dt

    if rank == 0:
        print(json.dumps({
            "metric": f"images/sec (whole node) {args.model} 224 "
                      f"{'bf16' if on_gpu else 'fp32'} train",
            "value": round(imgs_per_sec, 1),
            "unit": "img/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if on_gpu else "fp32",
            "data": "synthetic",
            "loss": round(float(loss), 4) if loss is not None else None,
            "config": {"model": args.model, "width_mult": args.width,
                       "global_batch": args.batch * world,
                       "image_size": 224, "parallelism": f"dp{world}",
                       "engine": args.engine, "graph": graphed is not None,
                       "optimizer": args.optimizer, "ema": args.ema,
                       "lr_schedule": ocfg.lr_schedule},
        }))
    udist.destroy()


if __name__ == "__main__":
    main()
