"""Mixed-workload stability soak: cycles renderers, hot-reloads materials/
emitters, saves/restores checkpoints, and asserts finite output each step —
the production-serving robustness check (run minutes-long on a GPU box;
also CPU-runnable at tiny sizes for CI).

Usage: python scripts/soak.py [--minutes 5] [--cpu] [--width 960] [--height 540]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=5.0)
    ap.add_argument("--cpu", action="store_true")
    ap.add_argument("--width", type=int, default=960)
    ap.add_argument("--height", type=int, default=540)
    args = ap.parse_args()

    import numpy as np
    import hippt
    from hippt.scene.procedural import cornell_box, kitchen, smoke_box

    dev = -1 if args.cpu else 0
    jobs = [
        ("kitchen-pt", kitchen(width=args.width, height=args.height, renderer="pt")),
        ("kitchen-wfpt", kitchen(width=args.width, height=args.height, renderer="wfpt")),
        ("smoke-vpt", smoke_box(width=args.width, height=args.height)),
        ("cornell-lt", cornell_box(width=args.width, height=args.height,
                                   renderer="lt", max_depth=5)),
    ]
    rs = [(name, hippt.PythonRenderer(d, device_id=dev)) for name, d in jobs]
    t_end = time.time() + args.minutes * 60.0
    it = 0
    spp = 1 if args.cpu else 8
    while time.time() < t_end:
        for name, r in rs:
            r.render(spp)
            img = r.render(spp)  # second call exercises warm accumulation
            arr = img.cpu().numpy() if hasattr(img, "cpu") else img
            assert np.isfinite(arr).all(), f"{name}: non-finite at iter {it}"
            assert arr[..., :3].mean() > 1e-4, f"{name}: black at iter {it}"
        # hot-reload churn on the kitchen (type switch + param update)
        from hippt.scene.scene import BsdfDesc, EmitterDesc
        _, rk = rs[0]
        rk.renderer.scene.set_bsdf(0, BsdfDesc(type="ggx", kg=(0.9, 0.7, 0.5),
                                               roughness_x=0.2, roughness_y=0.2))
        rk.renderer.scene.set_bsdf(0, BsdfDesc(type="lambertian", kd=(0.6, 0.6, 0.6)))
        rk.renderer.reset()
        # checkpoint round trip on the volumetric renderer
        _, rv = rs[2]
        ck = "/tmp/soak_ck.npz"
        rv.save_state(ck, step=it)
        extra = rv.load_state(ck)
        assert int(extra["step"]) == it
        it += 1
        if it % 4 == 0:
            print(f"[soak] iter {it}, {max(0.0, t_end - time.time()):.0f}s left, "
                  f"frames: " + " ".join(f"{n}={r.avg_frame_time():.0f}ms"
                                         for n, r in rs), flush=True)
    print(f"[soak] OK: {it} iterations, all renderers finite")


if __name__ == "__main__":
    main()
