"""hippt flagship benchmark — BASELINE.json metric:
Msamples/sec (whole node) on the 1080p modern-kitchen-class scene, sample-split
across N GPUs of one node with RCCL all-reduce (scripts/ddp_render.py parity).

Driver contract:
  python bench.py --gpus N --steps K --warmup W
N>1 is launched via torch.distributed.run (one rank per GPU over RCCL);
each rank renders `--spp-per-step` samples per pixel per step (weak scaling);
the radiance accumulators are spp-weighted all-reduced like the reference's
reduce_rendered_image (ddp_render.py:70-81).  Rank 0 prints ONE JSON line.
"""
import argparse
import json
import os
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=4)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--width", type=int, default=1920)
    ap.add_argument("--height", type=int, default=1080)
    ap.add_argument("--spp-per-step", type=int, default=32)
    ap.add_argument("--renderer", type=str, default=None,
                    help="pt | wfpt | pt-dyn (default: fastest available)")
    ap.add_argument("--scene", type=str, default="kitchen")
    ap.add_argument("--reduce-interval", type=int, default=1,
                    help="all-reduce the accumulator every this many steps")
    ap.add_argument("--cpu", action="store_true", help="CPU reference path (debug)")
    ap.add_argument("--sbvh", action="store_true", help="spatial-split BVH for the scene")
    args = ap.parse_args()

    import torch

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    distributed = world_size > 1

    backend = os.environ.get("HIPPT_DDP_BACKEND", "gloo" if args.cpu else "nccl")
    if distributed:
        import torch.distributed as dist
        dist.init_process_group(backend=backend)

    import hippt
    from hippt.scene import procedural

    renderer_kind = args.renderer or "pt"
    if args.scene == "kitchen":
        desc = procedural.kitchen(width=args.width, height=args.height,
                                  renderer=renderer_kind)
    elif args.scene == "cornell":
        desc = procedural.cornell_box(width=args.width, height=args.height,
                                      renderer=renderer_kind, max_depth=5)
    elif args.scene == "sports-car":
        desc = procedural.sports_car(width=args.width, height=args.height,
                                     renderer=renderer_kind)
    elif args.scene == "smoke":
        desc = procedural.smoke_box(width=args.width, height=args.height)
        renderer_kind = "vpt"
    else:
        raise SystemExit(f"unknown scene {args.scene}")
    if args.sbvh or os.environ.get("HIPPT_SBVH"):
        desc.config.use_sbvh = True

    # more ranks than GPUs (single-GPU RCCL/gloo smoke): share devices
    dev_id = local_rank if args.cpu else local_rank % max(1, torch.cuda.device_count())
    if not args.cpu:
        torch.cuda.set_device(dev_id)
    r = hippt.PythonRenderer(desc, device_id=-1 if args.cpu else dev_id,
                             seed_offset=rank)
    rend = r.renderer

    def barrier_sync():
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        if not args.cpu:
            torch.cuda.synchronize()

    def step():
        rend.render(args.spp_per_step)

    def reduce_accum():
        """spp-weighted all-reduce of the radiance accumulator (the reference's
        reduce_rendered_image: accumulators are sums, so SUM-reduce is exact)."""
        if not distributed:
            return
        import torch.distributed as dist
        t = rend.accum if not args.cpu else torch.from_numpy(rend.accum)
        if backend != "nccl" and hasattr(t, "is_cuda") and t.is_cuda:
            # gloo smoke (more ranks than GPUs): collective over a host copy
            h = t.cpu()
            dist.all_reduce(h, op=dist.ReduceOp.SUM)
            t.copy_(h.to(t.device))
        else:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
        # keep per-rank accumulators independent afterwards for weak scaling:
        # divide back so ranks continue from their own estimate
        t /= world_size

    # ---- warmup
    for _ in range(args.warmup):
        step()
    barrier_sync()

    # ---- timed region
    t0 = time.perf_counter()
    for k in range(args.steps):
        step()
        if (k + 1) % args.reduce_interval == 0:
            reduce_accum()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if distributed:
        import torch.distributed as dist
        te = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())

    n_gpus = world_size if distributed else (0 if args.cpu else 1)
    samples = float(args.width) * args.height * args.spp_per_step * args.steps * max(world_size, 1)
    msps = samples / elapsed / 1e6
    ms_per_step = elapsed * 1000.0 / args.steps

    if rank == 0:
        out = {
            "metric": "Msamples/sec (whole node), 1080p modern-kitchen",
            "value": round(msps, 3),
            "unit": "Msamples/s",
            "n_gpus": max(n_gpus, 1),
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic (procedural modern-kitchen-class scene, random-init materials)",
            "config": {
                "model": "modern-kitchen-proc",
                "global_batch": args.spp_per_step * max(world_size, 1),
                "seq_len": args.width * args.height,
                "width": args.width,
                "height": args.height,
                "spp_per_step_per_gpu": args.spp_per_step,
                "renderer": rend.kind,
                "n_prims": int(r.info()["n_prims"]),
                "parallelism": f"dp{max(world_size,1)} (sample-split, "
                               f"{'RCCL' if backend == 'nccl' else backend} all-reduce)",
            },
        }
        print(json.dumps(out), flush=True)

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
