#!/usr/bin/env python3
"""Flagship training-step benchmark (BASELINE.json contract).

Measures whole-job images/sec at 224^2, bf16, synthetic data, random-init
weights, full train step (forward + loss + backward + DDP gradient all-reduce
+ SGD update) on N GPUs of one node.

BASELINE.json's headline metric names BOTH ResNet-50 and ViT-B/16: the default
invocation (no --model flag) measures both back-to-back and emits ONE JSON
line whose value is the ResNet-50 images/s, with the ViT-B/16 number carried
in config.models. Pass --model to measure a single model.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import sys
import time

import torch

from deeplearning_amd.core.dist import (barrier, cleanup, get_rank,
                                        get_world_size, init_distributed)
from deeplearning_amd.data.synthetic import DeviceBatchLoader
from deeplearning_amd.models import build_model
from deeplearning_amd.ops import cross_entropy
from deeplearning_amd.parallel import wrap_data_parallel


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--model", default="all",
                   choices=["all", "resnet50", "vit_b16"],
                   help="'all' (default) = both headline models, "
                        "value = ResNet-50 img/s, ViT in config.models")
    p.add_argument("--batch-size", type=int, default=0,
                   help="per-GPU batch (0 = model default)")
    p.add_argument("--dp", default="bucketed", choices=["bucketed", "torch"])
    p.add_argument("--bucket-mb", type=float, default=64.0,
                   help="DDP gradient-bucket size sweep knob (16/32/64/128)")
    p.add_argument("--no-channels-last", action="store_true")
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--graph", default="off", choices=["auto", "on", "off"],
                   help="capture the train step in a hipGraph (single-GPU "
                        "only; removes per-kernel launch overhead)")
    return p.parse_args()


def run_model(args, model_name, device, world, rank, has_gpu):
    """Time one model's train loop; returns (images_per_sec, ms_per_step,
    global_batch) with elapsed maxed over ranks."""
    per_gpu_batch = args.batch_size
    if per_gpu_batch == 0:
        per_gpu_batch = 256 if has_gpu else 8

    torch.manual_seed(1234)
    model = build_model(model_name, num_classes=1000).to(device)
    channels_last = (model_name == "resnet50") and not args.no_channels_last \
        and has_gpu
    if channels_last:
        model = model.to(memory_format=torch.channels_last)
    model = wrap_data_parallel(model, style=args.dp, bucket_cap_mb=args.bucket_mb) \
        if world > 1 else model

    params = [p for p in model.parameters() if p.requires_grad]
    optimizer = torch.optim.SGD(params, lr=args.lr, momentum=0.9,
                                weight_decay=1e-4)

    loader = DeviceBatchLoader(per_gpu_batch, (3, 224, 224), 1000,
                               steps=args.warmup + args.steps, device=device,
                               distinct=2, seed=42 + rank,
                               channels_last=channels_last)
    batches = loader.batches
    amp_dtype = torch.bfloat16
    amp_enabled = has_gpu
    finalize = getattr(model, "finalize", None)

    use_graph = args.graph == "on" or (
        args.graph == "auto" and world == 1 and has_gpu)

    def one_step(i, set_to_none=True):
        x, y = batches[i % len(batches)]
        with torch.autocast("cuda", dtype=amp_dtype, enabled=amp_enabled):
            logits = model(x)
            loss = cross_entropy(logits, y)
        loss.backward()
        if finalize is not None:
            finalize()
        optimizer.step()
        optimizer.zero_grad(set_to_none=set_to_none)
        return loss

    # warmup (also absorbs MIOpen find); grads stay allocated for capture
    for i in range(args.warmup):
        one_step(i, set_to_none=not use_graph)

    graph = None
    if use_graph:
        # capture one full train step (fwd + loss + bwd + SGD) as a hipGraph;
        # replay re-executes every kernel on the same static batch buffers
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(2):
                    one_step(0, set_to_none=False)
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                one_step(0, set_to_none=False)
        except Exception as e:  # capture unsupported for this model: run eager
            print(f"[bench] graph capture failed ({e}); eager path",
                  file=sys.stderr)
            graph = None

    barrier()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    static_x, static_y = batches[0]
    for i in range(args.steps):
        if graph is not None:
            # rotate fresh data into the captured static buffers
            src_x, src_y = batches[(args.warmup + i) % len(batches)]
            if src_x.data_ptr() != static_x.data_ptr():
                static_x.copy_(src_x)
                static_y.copy_(src_y)
            graph.replay()
        else:
            one_step(args.warmup + i)
    barrier()
    if has_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # take the max elapsed over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if has_gpu else "cpu")
    if world > 1:
        import torch.distributed as dist

        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed_max = float(t.item())

    del graph, model, optimizer, loader, batches, static_x, static_y
    if has_gpu:
        torch.cuda.empty_cache()

    global_batch = per_gpu_batch * world
    images_per_sec = global_batch * args.steps / elapsed_max
    ms_per_step = elapsed_max / args.steps * 1000
    return images_per_sec, ms_per_step, global_batch


def main():
    args = parse_args()
    info = init_distributed()
    world = get_world_size()
    rank = get_rank()
    has_gpu = torch.cuda.is_available()
    device = torch.device("cuda", info["local_rank"]) if has_gpu else torch.device("cpu")
    if has_gpu:
        torch.cuda.set_device(device)
        torch.backends.cudnn.benchmark = True
    # rank-count evidence for the driver's SCALE record: every rank reports in
    print(f"[bench] rank {rank}/{world} local_rank {info['local_rank']} "
          f"device {device} backend "
          f"{'nccl(RCCL)' if has_gpu and world > 1 else ('gloo' if world > 1 else 'none')}",
          file=sys.stderr, flush=True)

    model_names = ["vit_b16", "resnet50"] if args.model == "all" else [args.model]
    results = {}
    for name in model_names:
        img_s, ms, global_batch = run_model(args, name, device, world, rank,
                                            has_gpu)
        results[name] = {"images_per_sec": round(img_s, 2),
                         "ms_per_step": round(ms, 3),
                         "global_batch": global_batch}

    primary = "resnet50" if "resnet50" in results else model_names[0]
    if rank == 0:
        r = results[primary]
        config = {
            "model": primary if len(results) == 1 else "resnet50 & vit_b16",
            "global_batch": r["global_batch"],
            "seq_len": None,
            "image_size": 224,
            "parallelism": f"dp{world}",
        }
        if len(results) > 1:
            config["models"] = results
        result = {
            "metric": "images/sec",
            "value": r["images_per_sec"],
            "unit": "images/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": r["ms_per_step"],
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if has_gpu else "fp32",
            "data": "synthetic",
            "config": config,
        }
        print(json.dumps(result), flush=True)
    cleanup()


if __name__ == "__main__":
    main()
