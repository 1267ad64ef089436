"""Batch/serial rendering driver: per-frame volume sequences and ToF sweeps.

Capability parity: reference scripts/serial_render.py — re-renders a scene per
VDB frame (smoke animation) or per ToF time window by updating the scene
parameters between frames, with dist.barrier() sync and per-frame PNG output
(serial_render.py:155-251).  Instead of rewriting XML on disk the jobs mutate
the SceneDesc directly.
"""
from __future__ import annotations

import argparse
import os
import time


def job_tof_rendering(args):
    """Sweep the ToF gating window across `frames` slices of path time
    (serial_render.py job_tof_rendering parity)."""
    import torch
    import torch.distributed as dist
    import hippt
    from ..scene import procedural
    from ..scene.xml_parser import parse_xml

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size > 1:
        dist.init_process_group(backend="gloo" if args.cpu else "nccl")

    desc = (parse_xml(args.scene) if args.scene.endswith(".xml")
            else procedural.cornell_box(width=args.width, height=args.height,
                                        renderer="vpt", max_depth=16))
    os.makedirs(args.outdir, exist_ok=True)
    window = args.time_window
    for f in range(args.frames):
        t0 = args.time_start + f * args.time_step
        desc.config.use_tof = True
        desc.config.min_time = t0
        desc.config.max_time = t0 + window
        r = hippt.PythonRenderer(desc, device_id=-1 if args.cpu else local_rank,
                                 seed_offset=rank)
        spp = max(1, args.spp // max(world_size, 1))
        r.render(spp=spp)
        if world_size > 1:
            from .ddp import reduce_rendered_image
            merged, _ = reduce_rendered_image(dist, r.renderer, world_size, cpu=args.cpu)
            dist.barrier()
        else:
            merged = None
        if rank == 0:
            path = os.path.join(args.outdir, f"tof_{f:04d}.png")
            if merged is not None:
                from ..utils.png import tonemap, write_png
                acc = merged.cpu().numpy()
                acc[:, :, 3] = 1.0
                write_png(path, tonemap(acc))
            else:
                r.save(path)
            print(f"[serial] frame {f} window [{t0:.2f},{t0+window:.2f}] -> {path}",
                  flush=True)
        r.release()
    if world_size > 1:
        dist.destroy_process_group()


def job_volume_sequence(args):
    """Render a sequence of procedural smoke frames (vdb-sequence parity)."""
    import torch.distributed as dist
    import hippt
    from ..scene import procedural

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world_size > 1:
        dist.init_process_group(backend="gloo" if args.cpu else "nccl")
    os.makedirs(args.outdir, exist_ok=True)
    for f in range(args.frames):
        desc = procedural.smoke_box(width=args.width, height=args.height)
        # evolve the plume: new noise seed per frame
        desc.media[0].density = procedural.smoke_density(n=64, seed=f) * 18.0
        r = hippt.PythonRenderer(desc, device_id=-1 if args.cpu else local_rank,
                                 seed_offset=rank)
        r.render(spp=max(1, args.spp // max(world_size, 1)))
        if world_size > 1:
            dist.barrier()
        if rank == 0:
            r.save(os.path.join(args.outdir, f"smoke_{f:04d}.png"))
            print(f"[serial] smoke frame {f} done", flush=True)
        r.release()
    if world_size > 1:
        dist.destroy_process_group()


def main(argv=None):
    ap = argparse.ArgumentParser("hippt.parallel.serial")
    ap.add_argument("--job", choices=["tof", "volume"], default="tof")
    ap.add_argument("--scene", type=str, default="cornell")
    ap.add_argument("--frames", type=int, default=8)
    ap.add_argument("--spp", type=int, default=64)
    ap.add_argument("--width", type=int, default=512)
    ap.add_argument("--height", type=int, default=512)
    ap.add_argument("--time-start", type=float, default=3.0)
    ap.add_argument("--time-step", type=float, default=0.5)
    ap.add_argument("--time-window", type=float, default=0.5)
    ap.add_argument("--outdir", type=str, default="serial_out")
    ap.add_argument("--cpu", action="store_true")
    args = ap.parse_args(argv)
    if args.job == "tof":
        job_tof_rendering(args)
    else:
        job_volume_sequence(args)


if __name__ == "__main__":
    main()
