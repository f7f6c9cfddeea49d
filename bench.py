#!/usr/bin/env python3
"""bench.py — flagship training-step benchmark (driver contract).

Measures the BASELINE.json headline: imgs/sec (whole-job aggregate) for the
CIFAR-100 B0-Inc10 ResNet-18 WA training step on MI355X — student fwd + frozen
teacher fwd (KD over the 90 known classes) + CE + KD losses + backward with
bucketed RCCL all-reduce overlap + fused SGD — bf16 compute, batch 128/GPU,
synthetic CIFAR-shaped data, random-init weights (no network for datasets).

Single GPU: python bench.py --gpus 1 --steps 30 --warmup 10
Multi-GPU (launched by the driver): python -m torch.distributed.run --nnodes=1
  --nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...
"""

import argparse
import json
import os
import time

import torch
import torch.distributed as dist

from cilfw import ops
from cilfw.distributed import DataParallelEngine
from cilfw.models import CilModel
from cilfw.optim import FlatSGD


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch_size", type=int, default=128)
    p.add_argument("--model", type=str, default="resnet18")
    p.add_argument("--input_size", type=int, default=32)
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--no_kd", action="store_true",
                   help="drop the teacher/KD part of the step")
    p.add_argument("--no_graph", dest="graph", action="store_false",
                   default=True,
                   help="disable hipGraph capture of the training step")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if use_cuda else "cpu"
    if use_cuda:
        torch.cuda.set_device(local_rank)
    if world > 1:
        dist.init_process_group(backend="nccl" if use_cuda else "gloo")

    dtype = torch.bfloat16 if (args.dtype == "bf16" and use_cuda) else torch.float32

    torch.manual_seed(1234 + rank)
    # B0-Inc10 mid-run shape: 10 heads x 10 classes, teacher knows 90
    model = CilModel(args.model, args.input_size).to(device)
    for _ in range(10):
        model.prev_model_adaption(10)
    model = model.to(device)
    known = 90
    teacher = None
    if not args.no_kd:
        teacher = model.copy()
        import torch.nn as nn
        teacher.fc.heads = nn.ModuleList(list(teacher.fc.heads)[:9])  # 90 classes
        teacher.freeze(["all"])
        teacher = teacher.to(device)
        if dtype == torch.bfloat16:
            teacher.cast_compute_weights_(dtype)

    engine = DataParallelEngine(model, bucket_mb=25.0)
    opt = FlatSGD(engine, lr=0.1, momentum=0.9, weight_decay=5e-4)

    B = args.batch_size
    x = torch.randn(B, args.input_size, args.input_size, 3, device=device,
                    dtype=dtype)
    y = torch.randint(0, 100, (B,), device=device)

    # the frozen teacher's forward is independent of the student's: run it on
    # a side HIP stream, overlapped with the student forward (fork/join — also
    # captured as a forked hipGraph)
    tstream = torch.cuda.Stream() if (use_cuda and teacher is not None) else None

    def step():
        opt.zero_grad()
        if tstream is not None:
            tstream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(tstream), torch.no_grad():
                t_logits, _ = teacher(x)
        logits, _ = model(x)
        if teacher is not None:
            if tstream is not None:
                torch.cuda.current_stream().wait_stream(tstream)
            else:
                with torch.no_grad():
                    t_logits, _ = teacher(x)
            loss, _lce, _lkd = ops.wa_loss(logits, t_logits, y, 0.0, 2.0, 0.5)
        else:
            loss, _lce, _lkd = ops.wa_loss(logits, None, y, 0.0, 2.0, 0.5)
        loss.backward()
        engine.finalize()
        opt.step()

    for _ in range(args.warmup):
        step()

    # hipGraph capture: the whole step (fwd + teacher fwd + losses + backward
    # incl. RCCL all-reduce + fused SGD) replays as ONE graph launch —
    # MI355X-idiomatic replacement for a tracing compiler.
    run_step = step
    graphed = False
    # multi-rank graph capture of RCCL collectives is untestable on the 1-GPU
    # dev boxes — default it OFF for world > 1 (eager measured within ~1% of
    # graphed); opt back in with CILFW_GRAPH_MULTI=1
    if world > 1 and os.environ.get("CILFW_GRAPH_MULTI") != "1":
        args.graph = False
    if args.graph and use_cuda:
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                step()
            torch.cuda.current_stream().wait_stream(s)
            gr = torch.cuda.CUDAGraph()
            with torch.cuda.graph(gr):
                step()
            run_step = gr.replay
            graphed = True
        except Exception as e:  # eager fallback, report it
            print(f"[bench] graph capture failed ({type(e).__name__}: {e}); "
                  f"running eager", flush=True)
            run_step = step

    for _ in range(3):
        run_step()

    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    if use_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    if world > 1:
        dist.barrier()
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000.0
        imgs_per_sec = B * world * args.steps / elapsed
        print(json.dumps({
            "metric": "imgs/sec/node (CIFAR-100 B0-Inc10 ResNet-18 WA+KD step)",
            "value": round(imgs_per_sec, 1),
            "unit": "imgs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {"model": args.model, "global_batch": B * world,
                       "input_size": args.input_size,
                       "kd": not args.no_kd, "hip_graph": graphed,
                       "parallelism": f"dp{world}"},
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
