"""Flagship benchmark: FlowNetS unsupervised training on FlyingChairs-
shaped synthetic data, 512x384 bf16, batch 64/GPU (BASELINE.json
configs[1]), 1 process per GPU over RCCL.

    python bench.py --gpus 1 --steps 20 --warmup 5
    torchrun --nproc-per-node 8 bench.py --gpus 8 ...

Prints ONE JSON line from rank 0: whole-job imgs/sec (max step time
over ranks), weak scaling.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--height", type=int, default=384)
    ap.add_argument("--width", type=int, default=512)
    ap.add_argument("--model", type=str, default="flownets")
    ap.add_argument("--dtype", type=str, default="bf16")
    ap.add_argument("--channels-last", action="store_true", default=True)
    ap.add_argument("--no-channels-last", dest="channels_last",
                    action="store_false")
    ap.add_argument("--graphs", action="store_true", default=False,
                    help="capture the train step in a hipGraph (no gain at\n"
                         "94%% GPU busy; kept as an option)")
    ap.add_argument("--no-graphs", dest="graphs", action="store_false")
    ap.add_argument("--bucket-mb", type=float, default=25.0,
                    help="DDP gradient bucket size (MB) for sweep runs")
    args = ap.parse_args()

    import torch.distributed as dist

    from deepof_amd.engine.optim import FusedAdam
    from deepof_amd.losses import MultiScaleUnsupLoss, preprocess_images
    from deepof_amd.losses.unsup import DATASET_MEANS
    from deepof_amd.models import build_model
    from deepof_amd.parallel import BucketedDataParallel

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    # --gpus must match the actual launch (torchrun --nproc-per-node N):
    # a silent mismatch would report the wrong whole-job aggregate
    if world != args.gpus:
        raise SystemExit(
            f"bench.py --gpus {args.gpus} but WORLD_SIZE={world}; launch "
            f"with torch.distributed.run --nproc-per-node {args.gpus}")
    # CPU/gloo fallback exists so the FULL distributed path (DDP wrapper,
    # barriers, reduction, JSON contract) is testable without a GPU
    use_cuda = torch.cuda.is_available()
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo")
    if use_cuda:
        torch.cuda.set_device(local_rank)
        dev = torch.device(f"cuda:{local_rank}")
    else:
        dev = torch.device("cpu")
    torch.manual_seed(1234 + rank)

    torch.backends.cudnn.benchmark = True  # MIOpen find for each conv shape
    model, flow_scales, weights = build_model(args.model)
    model.to(dev)
    if args.channels_last:
        model.to(memory_format=torch.channels_last)
    mean = DATASET_MEANS["flying_chairs"]
    loss_fn = MultiScaleUnsupLoss(flow_scales, weights, mean)
    if world > 1:
        model = BucketedDataParallel(model, bucket_cap_mb=args.bucket_mb)
    opt = FusedAdam(model.parameters(), lr=1.6e-5)

    # synthetic FlyingChairs-shaped data, resident on device; a few
    # distinct batches cycle so no step sees cached activations
    n_batches = 4
    batches = []
    for i in range(n_batches):
        img1 = torch.rand(args.batch, 3, args.height, args.width,
                          device=dev) * 255
        img2 = img1.roll(shifts=(2, -3), dims=(2, 3)) * 0.9 + \
            torch.rand_like(img1) * 0.1 * 255
        x = torch.cat([preprocess_images(img1, mean),
                       preprocess_images(img2, mean)], dim=1)
        if args.channels_last:
            x = x.to(memory_format=torch.channels_last)
        batches.append((x, img1, img2))

    use_bf16 = args.dtype == "bf16" and use_cuda

    def step(i):
        x, img1, img2 = batches[i % n_batches]
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16):
            flows = model(x)
        res = loss_fn(flows, img1, img2)
        res["total"].backward()
        if isinstance(model, BucketedDataParallel):
            model.finish_gradient_sync()
        opt.step()
        if isinstance(model, BucketedDataParallel):
            model.zero_grad_buckets()
        else:
            opt.zero_grad(set_to_none=False)
        return res["total"]

    # hipGraph capture: the whole fwd+loss+bwd+allreduce+Adam step is
    # recorded once and replayed per iteration; the input staging copy
    # (new batch -> static buffers) and the Adam hyper update (pinned
    # host read at replay) stay outside the graph.
    graph = None
    static = None
    adam_steps = args.warmup  # Adam state steps already taken

    def hyper_update(nstep):
        b1, b2 = 0.9, 0.999
        opt._hyper_pin[0] = 1.6e-5
        opt._hyper_pin[1] = 1.0 - b1**nstep
        opt._hyper_pin[2] = 1.0 - b2**nstep

    if args.graphs and world == 1 and use_cuda:
        try:
            for i in range(max(args.warmup, 3)):
                step(i)
            adam_steps = max(args.warmup, 3)
            static = tuple(t.clone() for t in batches[0])
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                x, img1, img2 = static
                with torch.autocast("cuda", dtype=torch.bfloat16,
                                    enabled=use_bf16):
                    flows = model(x)
                res = loss_fn(flows, img1, img2)
                res["total"].backward()
                opt.step()
                opt.zero_grad(set_to_none=False)
                graph_loss = res["total"]
            adam_steps += 1
        except Exception as e:  # pragma: no cover - capture unsupported
            print(f"# graph capture failed, falling back to eager: {e}",
                  file=sys.stderr)
            graph = None

    if graph is None:
        for i in range(args.warmup):
            step(i)

    def timed_step(i):
        nonlocal adam_steps
        if graph is not None:
            src = batches[i % n_batches]
            for dst, s_ in zip(static, src):
                dst.copy_(s_, non_blocking=True)
            adam_steps += 1
            hyper_update(adam_steps)
            graph.replay()
            return graph_loss
        return step(i)

    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    # per-step boundaries: CUDA events (async, ~µs each) on GPU, wall
    # clock on the CPU fallback -> p50/p90 distribution in the JSON
    if use_cuda:
        marks = [torch.cuda.Event(enable_timing=True)
                 for _ in range(args.steps + 1)]
    t0 = time.perf_counter()
    last = None
    cpu_marks = [t0]
    for i in range(args.steps):
        if use_cuda:
            marks[i].record()
        last = timed_step(i)
        if not use_cuda:
            cpu_marks.append(time.perf_counter())
    if use_cuda:
        marks[args.steps].record()
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t1 = time.perf_counter()

    if use_cuda:
        step_ms = [marks[i].elapsed_time(marks[i + 1])
                   for i in range(args.steps)]
    else:
        step_ms = [(cpu_marks[i + 1] - cpu_marks[i]) * 1000.0
                   for i in range(args.steps)]
    step_ms.sort()

    def pct(p):
        return step_ms[min(len(step_ms) - 1, int(p * len(step_ms)))]

    elapsed = t1 - t0
    if world > 1:
        t = torch.tensor([elapsed], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t)

    ms_per_step = elapsed / args.steps * 1000.0
    imgs_per_sec = args.batch * world * args.steps / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": f"imgs/sec {args.model}@FlyingChairs",
            "value": imgs_per_sec,
            "unit": "imgs/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "ms_p50": pct(0.50),
            "ms_p90": pct(0.90),
            "ms_min": step_ms[0],
            "ms_max": step_ms[-1],
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "loss_last": float(last.detach()),
            "config": {
                "model": args.model,
                "global_batch": args.batch * world,
                "seq_len": None,
                "image_size": [args.height, args.width],
                "parallelism": f"dp{world}",
            },
        }))
    if rank == 0 and os.environ.get("DEEPOF_DUMP_DISPATCH"):
        from deepof_amd.ops import conv as _c
        from deepof_amd.ops import deconv as _d

        for name, cache in (("conv", _c._dispatch_cache),
                            ("wrw", _c._wrw_cache),
                            ("deconv", _d._deconv_cache),
                            ("bwd2", _d._bwd_cache)):
            for k, v in cache.items():
                print(f"# dispatch {name}: {k} -> {v}", file=sys.stderr)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
