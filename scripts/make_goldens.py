"""(Re)generate the CPU golden images for tests/test_golden.py.
Run after an INTENTIONAL rendering-semantics change and commit the .npz."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np  # noqa: E402
import hippt  # noqa: E402
from hippt.scene.xml_parser import parse_xml  # noqa: E402

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
OUT = os.path.join(ROOT, "tests", "golden")

JOBS = [("cornell-box.xml", 48, 24), ("balls.xml", 64, 24),
        ("grid-cbox.xml", 40, 16), ("diamonds.xml", 48, 24),
        ("env-balls.xml", 64, 16), ("point-cbox.xml", 40, 24),
        ("hero.xml", 64, 16), ("grid-cbox-nvdb.xml", 40, 16),
        ("medium-cbox.xml", 40, 16), ("water-cbox.xml", 40, 16)]


def render(name, w, spp):
    d = parse_xml(os.path.join(ROOT, "scenes", name))
    d.camera.height = max(16, int(w * d.camera.height / d.camera.width))
    d.camera.width = w
    r = hippt.PythonRenderer(d, device_id=-1)
    return r.render(spp=spp).numpy().astype(np.float32)


def main():
    os.makedirs(OUT, exist_ok=True)
    for name, w, spp in JOBS:
        img = render(name, w, spp)
        np.savez_compressed(os.path.join(OUT, name.replace(".xml", "") + ".npz"),
                            img=img, spp=spp, w=w)
        print(name, img.shape, float(img[..., :3].mean()))
    # procedural flagship scenes (tiny CPU renders)
    import hippt
    from hippt.scene.procedural import kitchen, sports_car
    for gen, nm in [(kitchen, "kitchen"), (sports_car, "sports-car")]:
        d = gen(width=64, height=36)
        r = hippt.PythonRenderer(d, device_id=-1)
        img = r.render(spp=8).numpy().astype(np.float32)
        np.savez_compressed(os.path.join(OUT, nm + ".npz"), img=img, spp=8, w=64)
        print(nm, img.shape, float(img[..., :3].mean()))


if __name__ == "__main__":
    main()
