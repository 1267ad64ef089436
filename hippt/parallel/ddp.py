"""Multi-GPU sample-split DDP rendering over RCCL/xGMI.

Capability parity: reference scripts/ddp_render.py — one process per GPU via
torch.distributed (backend "nccl" = RCCL on ROCm), per-rank decorrelated RNG
seeds (local_rank*4201 + user seed), per-step 1-spp accumulation, and every
`reduce_interval` steps an spp-weighted all_reduce(SUM) of the radiance
accumulator (ddp_render.py:70-81), frame-time all_gather for logging
(:192-211), SIGINT graceful shutdown (:51-57).

MI355X-native notes: the all-reduce runs on our own accumulation tensors
(which are sums, so SUM-reduce is exact, no spp weighting error).  The
collective is issued with async_op=True — RCCL runs it on its internal
comm stream — and is only waited on AFTER the next step's render kernels
have been launched on the compute stream, so the xGMI transfer overlaps the
next frame's megakernel (start_reduce/finish_reduce below; an all-reduce of
a 1080p fp32 accumulator is ~33 MB and costs ~1 ms/GPU-pair on xGMI).
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import time

import numpy as np


def build_argparser():
    ap = argparse.ArgumentParser("hippt.parallel.ddp")
    ap.add_argument("--config", type=str, default=None, help="JSON/YAML config file")
    ap.add_argument("--scene", type=str, default="kitchen",
                    help="XML path or procedural name (cornell|kitchen|sports-car|smoke)")
    ap.add_argument("--width", type=int, default=None)
    ap.add_argument("--height", type=int, default=None)
    ap.add_argument("--renderer", type=str, default=None)
    ap.add_argument("--spp", type=int, default=1024, help="total spp across ranks")
    ap.add_argument("--spp-per-call", type=int, default=4)
    ap.add_argument("--reduce-interval", type=int, default=128)
    ap.add_argument("--parallelism", type=str, default="sample",
                    choices=["sample", "tile"],
                    help="sample = reference-style sample-split (each rank "
                         "renders the full frame at spp/N); tile = row-band "
                         "split (each rank renders h/N rows at full spp; "
                         "megakernel renderers only)")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--output", type=str, default="ddp_render.png")
    ap.add_argument("--logdir", type=str, default=None, help="TensorBoard logdir")
    ap.add_argument("--cpu", action="store_true", help="gloo/CPU path (tests)")
    ap.add_argument("--checkpoint", type=str, default=None,
                    help="periodic per-rank accumulator snapshot path "
                         "(rank id is appended); resumes from it if present")
    ap.add_argument("--checkpoint-interval", type=int, default=256,
                    help="steps between snapshots")
    ap.add_argument("--adaptive", action="store_true",
                    help="variance-guided per-pixel budgets after the first "
                         "step (megakernel renderers)")
    return ap


def load_scene(args):
    from ..scene import procedural
    if args.scene.endswith(".xml"):
        from ..scene.xml_parser import parse_xml
        desc = parse_xml(args.scene)
    else:
        kw = {}
        if args.width:
            kw["width"] = args.width
        if args.height:
            kw["height"] = args.height
        desc = {
            "cornell": procedural.cornell_box,
            "kitchen": procedural.kitchen,
            "sports-car": procedural.sports_car,
            "smoke": procedural.smoke_box,
        }[args.scene](**kw)
    if args.renderer:
        desc.config.renderer = args.renderer
    return desc


def start_reduce(dist, rend, cpu=False):
    """Snapshot the accumulator on the compute stream and launch the SUM
    all-reduce asynchronously (RCCL comm stream).  The returned work handle
    is waited on by finish_reduce — after the caller has already queued the
    next frame's kernels, so the xGMI collective overlaps them.  When the
    backend cannot take CUDA tensors (gloo smoke with more ranks than
    GPUs), the collective runs over a host copy."""
    import torch
    import torch.distributed as tdist
    a = rend.accum
    if isinstance(a, torch.Tensor):
        gloo = tdist.get_backend() == "gloo" if tdist.is_initialized() else cpu
        merged = a.cpu().clone() if (gloo and a.is_cuda) else a.clone()
    else:
        merged = torch.from_numpy(a.copy())
    work = dist.all_reduce(merged, op=dist.ReduceOp.SUM, async_op=True)
    return work, merged


def finish_reduce(work, merged):
    work.wait()
    cnt = merged[:, :, 3:4].clamp(min=1e-9)
    out = merged.clone()
    out[:, :, :3] /= cnt
    return out, float(merged[0, 0, 3])


def reduce_rendered_image(dist, rend, world_size, cpu=False):
    """Synchronous spp-weighted all-reduce (ddp_render.py:70-81).  Our
    accumulators are radiance SUMS with the sample count in alpha, so a plain
    SUM all-reduce is exactly the spp-weighted average; returns the merged
    mean image.  The main loop uses the split start_reduce/finish_reduce pair
    for comm/compute overlap; this wrapper serves tests and one-shot merges."""
    return finish_reduce(*start_reduce(dist, rend, cpu=cpu))


def main(argv=None):
    args = build_argparser().parse_args(argv)
    if args.config:
        import yaml
        with open(args.config) as f:
            cfg = yaml.safe_load(f)
        for k, v in cfg.items():
            setattr(args, k.replace("-", "_"), v)

    import torch
    import torch.distributed as dist

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    backend = os.environ.get("HIPPT_DDP_BACKEND", "gloo" if args.cpu else "nccl")
    if world_size > 1:
        dist.init_process_group(backend=backend)

    import hippt
    desc = load_scene(args)
    # per-rank decorrelated seeds (ddp_render.py:141-143)
    seed_offset = local_rank + args.seed * world_size
    dev_id = -1 if args.cpu else local_rank % max(1, torch.cuda.device_count())
    r = hippt.PythonRenderer(desc, device_id=dev_id, seed_offset=seed_offset)
    rend = r.renderer

    ckpt_path = f"{args.checkpoint}.rank{rank}.npz" if args.checkpoint else None
    ckpt_step = None
    if ckpt_path and os.path.exists(ckpt_path):
        extra = r.load_state(ckpt_path)
        # the step index is persisted explicitly (deriving it from accum_cnt
        # over-counts under --adaptive, whose per-call spp can exceed
        # spp_per_call)
        if extra and "step" in extra:
            ckpt_step = int(extra["step"])
        if rank == 0:
            print(f"[ddp] resumed from {ckpt_path} at {r.counter()} spp", flush=True)

    stop = {"flag": False}
    if rank == 0:
        def _sigint(sig, frame):
            stop["flag"] = True
        signal.signal(signal.SIGINT, _sigint)

    writer = None
    if args.logdir and rank == 0:
        try:
            from torch.utils.tensorboard import SummaryWriter
            writer = SummaryWriter(args.logdir)
        except Exception:
            writer = None

    if args.parallelism == "tile":
        # row-band split on 16-row tile boundaries; SUM all-reduce still
        # merges exactly (other ranks' rows are zero)
        h = desc.camera.height
        tiles = (h + 15) // 16
        t0_, t1_ = tiles * rank // world_size, tiles * (rank + 1) // world_size
        band = (t0_ * 16, min(h, t1_ * 16))
        spp_per_rank = args.spp
    else:
        band = (0, 0)
        spp_per_rank = max(1, args.spp // max(world_size, 1))
    steps = (spp_per_rank + args.spp_per_call - 1) // args.spp_per_call
    t_start = time.perf_counter()
    merged = None
    pending = None  # (work, merged_tensor, step_idx) of an in-flight reduce
    if ckpt_step is not None:
        start_step = min(ckpt_step, steps)
    else:
        start_step = min(rend.accum_cnt // args.spp_per_call, steps)

    def _finish_pending(pending):
        """Complete an async reduce (after the NEXT step's kernels were
        launched — this is where the xGMI collective overlaps compute) and
        do the frame-time all_gather + logging (ddp_render.py:192-211)."""
        work, m, kidx = pending
        merged, total_spp = finish_reduce(work, m)
        ft = torch.tensor([rend.avg_frame_time()])
        fts = [torch.zeros_like(ft) for _ in range(world_size)]
        dist.all_gather(fts, ft)
        if rank == 0:
            times = [float(t.item()) for t in fts]
            print(f"[ddp] step {kidx+1}/{steps} total_spp={total_spp:.0f} "
                  f"frame_ms={times} avg={np.mean(times):.1f}", flush=True)
            if writer is not None:
                img = (merged[:, :, :3].clamp(min=0) ** (1 / 2.1)).clamp(max=1)
                writer.add_image("render", img.permute(2, 0, 1).cpu(), kidx)
                writer.add_scalar("frame_ms/avg", float(np.mean(times)), kidx)
        return merged

    for k in range(start_step, steps):
        if args.adaptive and rend.accum_cnt >= args.spp_per_call and rend.rid in (0, 2):
            m = rend._spp_budget(args.spp_per_call)
            rend.render(args.spp_per_call, y0=band[0], y1=band[1], spp_map=m)
        else:
            rend.render(args.spp_per_call, y0=band[0], y1=band[1])
        if pending is not None:
            # this step's kernels are queued; now drain last step's collective
            merged = _finish_pending(pending)
            pending = None
        if ckpt_path and (k + 1) % args.checkpoint_interval == 0:
            r.save_state(ckpt_path, step=np.int64(k + 1))
        if world_size > 1 and ((k + 1) % args.reduce_interval == 0 or k == steps - 1):
            pending = (*start_reduce(dist, rend, cpu=args.cpu), k)
        if stop["flag"]:
            break
    if pending is not None:
        merged = _finish_pending(pending)
        pending = None

    if world_size > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t_start
    if merged is None:
        merged = torch.from_numpy(rend.raw()) if args.cpu else rend.raw()

    if rank == 0:
        total_samples = desc.camera.width * desc.camera.height * spp_per_rank * max(world_size, 1)
        msps = total_samples / elapsed / 1e6
        print(json.dumps({"whole_node_msamples_per_sec": round(msps, 2),
                          "elapsed_s": round(elapsed, 3),
                          "world_size": world_size,
                          "spp_per_rank": spp_per_rank}), flush=True)
        if args.output:
            from ..utils.png import tonemap, write_png
            acc = merged.cpu().numpy()
            acc[:, :, 3] = 1.0  # merged is already a mean image
            write_png(args.output, tonemap(acc))
    r.release()
    if world_size > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
