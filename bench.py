#!/usr/bin/env python3
"""Benchmark harness for the MI355X-native IMHN pose framework.

Measures the flagship 4-stage IMHN @512x512 bf16 on synthetic data with
random-init weights (no network access in this environment):

  * --mode infer (default): forward-only inference, batch 4, last-stack output
    — the reference's HEADLINE configuration (38.5 FPS on a 2080 Ti,
    BASELINE.md / reference test_inference_speed.py). Metric: FPS @512x512,
    vs_baseline = FPS / 38.5.
  * --mode train: full training step — synthetic batch -> forward -> focal-L2
    loss -> backward (with overlapped RCCL all-reduce when N > 1) -> fused SGD
    step. Metric: aggregate train_images_per_sec over all ranks (weak scaling:
    fixed per-GPU batch). --data device-gt additionally runs the on-device
    ground-truth generator inside the timed region (the full 512^2 pipeline
    stays on the GPU).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
         --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints exactly one JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from improved_body_parts_amd.config import GetConfig, TrainingOpt  # noqa: E402
from improved_body_parts_amd.data import SyntheticPoseDataset  # noqa: E402
from improved_body_parts_amd.engine import FusedSGD  # noqa: E402
from improved_body_parts_amd.models import Network, NetworkEval  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--mode", choices=["train", "infer"], default="infer")
    p.add_argument("--data", choices=["prestaged", "device-gt"], default="prestaged",
                   help="train-mode input: pre-staged batches (compute-only "
                        "timing) or the on-device GT generator inside the "
                        "timed region")
    p.add_argument("--batch", type=int, default=None,
                   help="per-GPU batch (default: 16 train / 4 infer)")
    p.add_argument("--nstack", type=int, default=4)
    p.add_argument("--input", type=int, default=512)
    p.add_argument("--no-bf16", action="store_true")
    p.add_argument("--graph", action="store_true",
                   help="capture the inference forward in a hipGraph and replay")
    p.add_argument("--allow-eager", action="store_true",
                   help="permit eager fallback if the HIP extension is absent")
    return p.parse_args()


def main():
    args = parse_args()
    if args.allow_eager:
        os.environ["IBP_AMD_ALLOW_EAGER"] = "1"

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world_size > 1
    use_cuda = torch.cuda.is_available()
    # modulo so a 2-rank run on a 1-GPU box (RCCL wiring validation) maps both
    # ranks onto the existing device instead of an invalid ordinal
    device = torch.device("cuda", local_rank % max(torch.cuda.device_count(), 1)) \
        if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if distributed:
        import torch.distributed as dist
        dist.init_process_group(backend="nccl" if use_cuda else "gloo",
                                init_method="env://")

    batch = args.batch or (16 if args.mode == "train" else 4)
    bf16 = use_cuda and not args.no_bf16
    dtype = torch.bfloat16 if bf16 else torch.float32

    cfg_name = {512: "Canonical", 384: "Canonical384", 768: "Canonical768"}.get(
        args.input, "Canonical")
    config = GetConfig(cfg_name)
    opt = TrainingOpt(nstack=args.nstack, batch_size=batch,
                      nstack_weight=[1] * args.nstack)

    torch.manual_seed(1234)  # same random init on all ranks

    if args.mode == "train":
        model = Network(opt, config, bn=True, dist=True).to(device)
        if bf16:
            model = model.bfloat16()
            for m in model.modules():
                if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                    m.float()
        model.train()
        optimizer = FusedSGD(model.parameters(), lr=opt.learning_rate * world_size,
                             momentum=opt.momentum, weight_decay=opt.weight_decay)
        reducer = None
        if distributed:
            from improved_body_parts_amd.parallel import GradReducer
            reducer = GradReducer(model, broadcast_parameters=True)

        if args.data == "device-gt" and use_cuda:
            # the on-device GT pipeline (HIP heatmapper + masks) runs INSIDE
            # the timed region — what a real training step pays
            from improved_body_parts_amd.data import DeviceGTSyntheticLoader
            loader = DeviceGTSyntheticLoader(
                config, batch, steps_per_epoch=1 << 30, seed=17 + rank,
                device=device, dtype=dtype)
            batch_iter = iter(loader)

            def get_batch(i):
                return next(batch_iter)
        else:
            # pre-stage a few synthetic batches on device (different per rank)
            ds = SyntheticPoseDataset(config, length=world_size * 2, seed=17)
            batches = []
            for b in range(2):
                idx = rank * 2 + b
                img, mm, hm = ds[idx]
                img = img[None].expand(batch, -1, -1, -1).contiguous()
                mm = mm[None].expand(batch, -1, -1, -1).contiguous()
                hm = hm[None].expand(batch, -1, -1, -1).contiguous()
                batches.append(tuple(t.to(device=device, dtype=dtype)
                                     for t in (img, mm, hm)))

            def get_batch(i):
                return batches[i % len(batches)]

        def train_step(b):
            if reducer is not None:
                reducer.zero_grad()
            else:
                # None grads skip ~800 fill launches AND turn the first
                # autograd accumulation per tensor into an assignment.
                # Graph mode needs STATIC grad buffers instead: with stable
                # addresses the fused-SGD pointer table never rebuilds, so
                # nothing capture-hostile runs inside the capture.
                optimizer.zero_grad(set_to_none=not args.graph)
            loss = model(b)
            loss.backward()
            if reducer is not None:
                reducer.finalize()
            optimizer.step()
            return loss

        if args.graph and args.data == "device-gt":
            raise SystemExit("--graph requires pre-staged data (--data prestaged)")
        if args.graph and use_cuda and not distributed:
            # whole-step hipGraph: fwd + focal-L2 + bwd + fused SGD captured
            # once, replayed per step (there is no host sync inside the step)
            static = batches[0]
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    train_step(static)
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                static_loss = train_step(static)

            def step(i):
                graph.replay()
                return static_loss
        else:
            def step(i):
                return train_step(get_batch(i))

        metric_name = "train_images_per_sec"
        vs_baseline = None
        higher = True
    else:
        model = NetworkEval(opt, config, bn=True).to(device)
        if bf16:
            model = model.bfloat16()
            for m in model.modules():
                if isinstance(m, torch.nn.modules.batchnorm._BatchNorm):
                    m.float()
        model.eval()
        img = torch.rand(batch, config.height, config.width, 3,
                         device=device, dtype=dtype)

        if args.graph and use_cuda:
            # hipGraph capture: one replay per step, zero per-kernel launch
            # overhead from the host
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s), torch.no_grad():
                for _ in range(3):
                    model(img)
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph), torch.no_grad():
                static_out = model(img)

            def step(i):
                graph.replay()
                return static_out[-1][0]
        else:
            @torch.no_grad()
            def step(i):
                out = model(img)
                return out[-1][0]  # last stack, scale 0 (reference evaluate.py:126)

        metric_name = "fps_512_infer"
        vs_baseline = None  # filled below from the 38.5 FPS headline
        higher = True

    def sync():
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        step(i)
    sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    sync()
    elapsed = time.perf_counter() - t0

    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=device if use_cuda else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t)

    if os.environ.get("IBP_DDP_TIMING") == "1" and args.mode == "train" \
            and use_cuda and distributed:
        # untimed extra steps measuring exposed vs total all-reduce time
        # (stderr only — stdout stays the single JSON contract line)
        step(0)
        step(1)
        reducer_t = reducer.last_timing if reducer is not None else None
        if rank == 0 and reducer_t:
            print(f"[ddp-timing] {reducer_t}", file=sys.stderr, flush=True)

    images = batch * args.steps * world_size
    value = images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    if args.mode == "infer":
        vs_baseline = value / 38.5  # reference headline: 38.5 FPS @512^2 batch 4

    if rank == 0:
        print(json.dumps({
            "metric": metric_name,
            "value": round(value, 2),
            "unit": "images/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": higher,
            "scaling": "weak",
            "vs_baseline": round(vs_baseline, 3) if vs_baseline is not None else None,
            "dtype": "bf16" if bf16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": f"{args.nstack}-stage IMHN @{args.input}x{args.input}",
                "mode": args.mode,
                # device-gt needs a GPU; report what actually ran
                "train_data": (args.data if use_cuda else "prestaged")
                if args.mode == "train" else None,
                "global_batch": batch * world_size,
                "input": args.input,
                "parallelism": f"dp{world_size}",
                "max_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 2)
                if use_cuda else None,
            },
        }), flush=True)

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
