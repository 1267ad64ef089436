"""Measure the adaptive-sampling payoff: equal total samples, compare the
image-wide and worst-tile variance of uniform vs variance-guided allocation."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np  # noqa: E402
import hippt  # noqa: E402
from hippt.scene.procedural import cornell_box  # noqa: E402


def run(adaptive, spp, dev, seed=0, w=192):
    d = cornell_box(width=w, height=w, spp=1, max_depth=6)
    r = hippt.PythonRenderer(d, device_id=dev, seed_offset=seed)
    r.render(spp=spp, adaptive=adaptive)
    out = r.renderer.raw()
    out = out.cpu().numpy() if hasattr(out, "cpu") else np.asarray(out)
    return out[..., :3]


def main():
    dev = 0 if "--gpu" in sys.argv else -1
    spp = int(sys.argv[sys.argv.index("--spp") + 1]) if "--spp" in sys.argv else 64
    w = 192 if dev >= 0 else 96
    ref = run(False, 32 * spp if dev >= 0 else 1536, dev, seed=7, w=w)
    u = run(False, spp, dev, w=w)
    a = run(True, spp, dev, w=w)
    ru = float(np.sqrt(((u - ref) ** 2).mean()))
    ra = float(np.sqrt(((a - ref) ** 2).mean()))
    print(f"uniform  RMSE vs converged ref: {ru:.4f}")
    print(f"adaptive RMSE vs converged ref: {ra:.4f}")
    print(f"error ratio {ru / ra:.2f}x  (~{(ru / ra) ** 2:.2f}x sample efficiency)")


if __name__ == "__main__":
    main()
