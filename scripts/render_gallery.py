"""Render every example scene to PNG (docs gallery / visual regression).
Usage: python scripts/render_gallery.py [outdir] [--cpu] [--small]"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import hippt  # noqa: E402
from hippt.scene.xml_parser import parse_xml  # noqa: E402
from hippt.scene.procedural import kitchen, sports_car  # noqa: E402

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    outdir = sys.argv[1] if len(sys.argv) > 1 and not sys.argv[1].startswith("-") else "gallery"
    cpu = "--cpu" in sys.argv
    small = "--small" in sys.argv
    dev = -1 if cpu else 0
    os.makedirs(outdir, exist_ok=True)
    hq = "--hq" in sys.argv
    m = 8 if hq else 1
    xml_jobs = [("cornell-box.xml", 512 * m), ("balls.xml", 512 * m), ("grid-cbox.xml", 256 * m),
                ("diamonds.xml", 512 * m), ("env-balls.xml", 256 * m), ("caustics-lt.xml", 512 * m),
                ("tof-cbox.xml", 384 * m), ("point-cbox.xml", 256 * m),
                ("dof-balls.xml", 256 * m), ("spot-cbox.xml", 256 * m),
                ("hero.xml", 384 * m), ("grid-cbox-nvdb.xml", 256 * m),
                ("medium-cbox.xml", 256 * m), ("water-cbox.xml", 256 * m)]
    for name, spp in xml_jobs:
        d = parse_xml(os.path.join(ROOT, "scenes", name))
        if small:
            d.camera.width //= 4
            d.camera.height //= 4
            spp = max(4, spp // 16)
        r = hippt.PythonRenderer(d, device_id=dev)
        r.render(spp=spp)
        out = os.path.join(outdir, name.replace(".xml", "") + ".png")
        r.save(out)
        print(f"{name} -> {out} ({r.avg_frame_time():.0f} ms/frame)", flush=True)
        r.release()
    for gen, nm, spp in [(kitchen, "kitchen", 256 * m), (sports_car, "sports-car", 256 * m)]:
        w, h = (240, 135) if small else (960, 540)
        d = gen(width=w, height=h)
        r = hippt.PythonRenderer(d, device_id=dev)
        r.render(spp=max(2, spp // 16) if small else spp)
        r.save(os.path.join(outdir, nm + ".png"))
        print(nm, flush=True)
        r.release()


if __name__ == "__main__":
    main()
