"""XML scene parsing -> render pipeline (parser parity with the reference
grammar: brdf/emitter/shape/sensor/renderer/accelerator elements)."""
import os

import numpy as np

import hippt
from hippt.scene.xml_parser import parse_xml, parse_rgb

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_parse_rgb():
    assert parse_rgb("#FF0000") == (1.0, 0.0, 0.0)
    assert parse_rgb("0.5") == (0.5, 0.5, 0.5)
    assert parse_rgb("1, 2, 3") == (1.0, 2.0, 3.0)


def test_cornell_xml_matches_procedural():
    d = parse_xml(os.path.join(ROOT, "scenes", "cornell-box.xml"))
    assert d.config.renderer == "pt"
    assert d.config.spp == 64 and d.config.max_depth == 5
    assert d.config.overlap_w == 0.6
    assert len(d.bsdfs) == 4 and len(d.emitters) == 1 and len(d.objects) == 8
    assert d.camera.width == 256
    d.camera.width = d.camera.height = 64
    r = hippt.PythonRenderer(d, device_id=-1)
    img = r.render(spp=16).numpy()
    # same structure as the procedural cornell box
    from hippt.scene.procedural import cornell_box
    d2 = cornell_box(width=64, height=64, spp=8, max_depth=5)
    img2 = hippt.PythonRenderer(d2, device_id=-1).render(spp=16).numpy()
    m1, m2 = img[..., :3].mean(), img2[..., :3].mean()
    assert abs(m1 - m2) / m2 < 0.03, (m1, m2)


def test_balls_xml_all_bsdf_types():
    d = parse_xml(os.path.join(ROOT, "scenes", "balls.xml"))
    types = {b.type for b in d.bsdfs}
    assert {"lambertian", "ggx", "glass", "dispersion", "plastic", "specular"} <= types
    etypes = {e.type for e in d.emitters}
    assert {"area", "point"} <= etypes
    # sphere primitives present
    assert any(o.spheres is not None for o in d.objects)
    d.camera.width, d.camera.height = 80, 45
    img = hippt.PythonRenderer(d, device_id=-1).render(spp=8).numpy()
    assert np.isfinite(img).all()
    assert img[..., :3].mean() > 0.01
