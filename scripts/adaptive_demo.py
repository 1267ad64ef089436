"""Measure the adaptive-sampling payoff: equal total samples, compare the
image-wide and worst-tile variance of uniform vs variance-guided allocation."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np  # noqa: E402
import hippt  # noqa: E402
from hippt.scene.procedural import cornell_box  # noqa: E402


def run(adaptive, spp, dev):
    d = cornell_box(width=256, height=256, spp=1, max_depth=6)
    r = hippt.PythonRenderer(d, device_id=dev)
    r.render(spp=spp, adaptive=adaptive)
    v = r.variance()
    v = v.cpu().numpy() if hasattr(v, "cpu") else np.asarray(v)
    v = v[..., 0]
    # tile-max: worst 16x16 tile mean variance (what the eye sees as noise)
    t = v.reshape(16, 16, 16, 16).mean(axis=(1, 3))
    return float(v.mean()), float(t.max())


def main():
    dev = 0 if "--gpu" in sys.argv else -1
    spp = int(sys.argv[sys.argv.index("--spp") + 1]) if "--spp" in sys.argv else 64
    mu, tu = run(False, spp, dev)
    ma, ta = run(True, spp, dev)
    print(f"uniform : mean var {mu:.3e}  worst-tile {tu:.3e}")
    print(f"adaptive: mean var {ma:.3e}  worst-tile {ta:.3e}")
    print(f"worst-tile variance ratio (uniform/adaptive): {tu / max(ta, 1e-30):.2f}x")


if __name__ == "__main__":
    main()
