"""Driver benchmark contract: flagship serving step on N GPUs of one node.

Measures the BASELINE.json headline metric: images/sec (whole node) for
ResNet-50 224x224 bf16, batch=64 per GPU (weak scaling), synthetic
images, random-init weights, through the native HIP-kernel engine
(captured HIP graph per batch size).

  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches via torch.distributed.run (one rank per GPU,
RCCL); rank 0 prints ONE JSON line.

`--device cpu` is a rehearsal mode (gloo backend, fp32 CPU executor) so
the torchrun rendezvous / barrier / all-reduce path can be exercised at
world>1 on machines with no GPU; it changes the backend, never the
measurement protocol.
"""
import argparse
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

# default steps sized so the 1-GPU ResNet-50 timed region is >= 1 s
# (~2.1 ms/step measured round 1) — a region the driver's SMI sampler
# can actually see (VERDICT round-1 weak #6).
DEFAULT_STEPS = 500
DEFAULT_WARMUP = 20


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=DEFAULT_STEPS)
    p.add_argument("--warmup", type=int, default=DEFAULT_WARMUP)
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--model", default="resnet50")
    p.add_argument("--device", default="cuda",
                   choices=["cuda", "cpu"],
                   help="cpu = gloo rehearsal mode (no GPU needed)")
    args = p.parse_args()

    import numpy as np
    import torch

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    on_gpu = args.device == "cuda"
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        if on_gpu:
            torch.cuda.set_device(local_rank)
        # nccl IS RCCL on ROCm; gloo for the CPU rehearsal
        dist.init_process_group(backend="nccl" if on_gpu else "gloo")

    if on_gpu:
        assert torch.cuda.is_available(), "bench needs a GPU"
        device = f"cuda:{local_rank}"
        torch.cuda.set_device(device)
    else:
        device = "cpu"

    from tensorflow_web_deploy_amd.conf.config import Config
    from tensorflow_web_deploy_amd.engine.service import LocalEngine
    if on_gpu:
        from tensorflow_web_deploy_amd.ops.native import \
            require_native_on_gpu
        require_native_on_gpu()

    shapes = {"lenet": (28, 1, 10)}
    size, chans, ncls = shapes.get(
        args.model, (299 if args.model == "inception_v3" else 224, 3, 1000))
    cfg = Config(model_name=args.model, device=device, input_height=size,
                 input_width=size, input_channels=chans, num_classes=ncls)
    eng = LocalEngine(cfg)

    def sync():
        if on_gpu:
            torch.cuda.synchronize()

    # synthetic images of the benchmark shape (no network: BASELINE.json)
    rng = np.random.default_rng(1234 + rank)
    x = rng.random((args.batch, size, size, chans), dtype=np.float32)
    xt = torch.from_numpy(x).to(device)

    feeds = {cfg.input_name: xt}

    # warmup (also compiles + captures the HIP graph)
    for _ in range(args.warmup):
        eng.ex.run(feeds)
    sync()
    if dist:
        dist.barrier()
        sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        eng.ex.run(feeds)
    sync()
    if dist:
        dist.barrier()
        sync()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if dist:
        t = torch.tensor([elapsed],
                         device=device if on_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    total_images = args.batch * args.steps * world
    ips = total_images / elapsed

    # no reference-published numbers exist (empty reference
    # tree, "published": {}); the round-1 measured numbers are the
    # self-set baseline of record (1x MI355X, b64, bf16)
    ROUND1_BASELINE = {"resnet50": 30451.7, "inception_v3": 19611.0,
                       "vit_b16": 11118.0, "mobilenet_v2": 50971.0}
    base = ROUND1_BASELINE.get(args.model) if on_gpu and world == 1 else None
    vs_base = round(ips / base, 4) if base else None

    if rank == 0:
        print(json.dumps({
            "metric": "images_per_sec",
            "value": round(ips, 2),
            "unit": "images/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": vs_base,
            "dtype": "bf16" if on_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * world,
                "image_size": size,
                "parallelism": f"dp{world}",
            },
        }))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
