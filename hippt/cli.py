"""Offline render CLI.

Capability parity: reference `pt` binary (app/pt_renderer.cu:26-117): parse
scene XML -> pick renderer (megakernel PT / WFPT / LT / VPT / PT-dynamic /
depth / BVH-cost) -> render spp -> write PNG.

Usage:
    python -m hippt.cli scene.xml [-o render.png] [--spp N] [--device 0|-1]
    python -m hippt.cli cornell --spp 64      (procedural scene names work too)
"""
from __future__ import annotations

import argparse
import sys
import time


def main(argv=None):
    ap = argparse.ArgumentParser("hippt.cli", description=__doc__)
    ap.add_argument("scene", help="scene XML path or procedural name "
                                  "(cornell|kitchen|sports-car|smoke)")
    ap.add_argument("-o", "--output", default="render.png")
    ap.add_argument("--spp", type=int, default=None, help="override sample count")
    ap.add_argument("--renderer", default=None,
                    help="pt|pt-dyn|wfpt|vpt|lt|bdpt|depth|bvh-cost")
    ap.add_argument("--device", type=int, default=None,
                    help="GPU id (default: 0 if available else CPU)")
    ap.add_argument("--width", type=int, default=None)
    ap.add_argument("--height", type=int, default=None)
    ap.add_argument("--gamma", type=float, default=2.1)
    ap.add_argument("--exposure", type=float, default=1.0,
                    help="radiance scale before gamma (bright scenes)")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--colormap", default="plasma", help="depth/bvh-cost false color")
    ap.add_argument("--variance", default=None, help="also write the variance map here")
    ap.add_argument("--adaptive", action="store_true",
                    help="variance-guided per-pixel sample allocation")
    ap.add_argument("--denoise", action="store_true",
                    help="SVGF-lite AOV-guided denoise of the final image")
    args = ap.parse_args(argv)

    import hippt
    if args.scene.endswith(".xml"):
        from .scene.xml_parser import parse_xml
        desc = parse_xml(args.scene)
    else:
        from .scene import procedural
        gens = {"cornell": procedural.cornell_box, "kitchen": procedural.kitchen,
                "sports-car": procedural.sports_car, "smoke": procedural.smoke_box}
        if args.scene not in gens:
            ap.error(f"unknown scene {args.scene}")
        desc = gens[args.scene]()
        desc.config.max_depth = max(desc.config.max_depth, 5)
    if args.width:
        desc.camera.width = args.width
    if args.height:
        desc.camera.height = args.height
    if args.renderer:
        desc.config.renderer = args.renderer
    spp = args.spp or desc.config.spp

    device = args.device
    if device is None:
        device = 0 if hippt.has_gpu() else -1
    print(f"[hippt] scene={args.scene} renderer={desc.config.renderer} "
          f"{desc.camera.width}x{desc.camera.height} spp={spp} "
          f"device={'cpu' if device < 0 else device}")
    t0 = time.perf_counter()
    r = hippt.PythonRenderer(desc, device_id=device, seed_offset=args.seed)
    if args.denoise and desc.config.renderer in ("pt", "pt-dyn", "vpt"):
        r.renderer.enable_aov()
    info = r.info()
    print(f"[hippt] prims={info['n_prims']} nodes={info['n_nodes']} "
          f"bsdfs={info['n_bsdfs']} emitters={info['n_emitters']} "
          f"sah={info.get('sah_cost', 0):.1f}")
    chunk = 16 if device >= 0 else 4
    done = 0
    while done < spp:
        step = min(chunk, spp - done)
        if args.adaptive and done >= chunk and desc.config.renderer in ("pt", "pt-dyn", "vpt"):
            r.renderer.render(step, spp_map=r.renderer._spp_budget(step))
        else:
            r.renderer.render(step)
        done += step
        el = time.perf_counter() - t0
        print(f"\r[hippt] {done}/{spp} spp, {el:.1f}s, "
              f"{r.avg_frame_time():.1f} ms/frame", end="", flush=True)
    print()

    if desc.config.renderer in ("depth", "bvh-cost"):
        import numpy as np
        from .utils.colormap import false_color
        from .utils.png import write_png
        acc = (r.renderer.accum.cpu().numpy() if r.renderer.device is not None
               else r.renderer.accum)
        vals = acc[:, :, 0] / np.maximum(acc[:, :, 3], 1e-9)
        write_png(args.output, false_color(vals, cmap=args.colormap,
                                           log_scale=desc.config.renderer == "bvh-cost"))
    elif args.denoise and getattr(r.renderer, "aux", None) is not None:
        import numpy as np
        from .utils.png import write_png, tonemap
        den = r.renderer.denoise()
        den = den.cpu().numpy() if hasattr(den, "cpu") else np.asarray(den)
        acc = np.concatenate([den, np.ones_like(den[..., :1])], axis=2)
        write_png(args.output, tonemap(acc, gamma=args.gamma))
    else:
        r.renderer.save(args.output, gamma=args.gamma, exposure=args.exposure)
    print(f"[hippt] wrote {args.output} ({time.perf_counter() - t0:.1f}s total)")

    if args.variance:
        import numpy as np
        from .utils.colormap import false_color
        from .utils.png import write_png
        v = r.variance()
        v = v.cpu().numpy() if hasattr(v, "cpu") else v
        write_png(args.variance, false_color(v[:, :, 0], cmap="viridis", log_scale=True))
        print(f"[hippt] wrote {args.variance}")
    r.release()


if __name__ == "__main__":
    main()
