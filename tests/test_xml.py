"""XML scene parsing -> render pipeline (parser parity with the reference
grammar: brdf/emitter/shape/sensor/renderer/accelerator elements)."""
import os

import numpy as np
import pytest

import hippt
from hippt.scene.xml_parser import parse_xml, parse_rgb

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SCENES = os.path.join(ROOT, "scenes")


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


class TestExampleScenes:
    """Every shipped example XML parses and renders finite, non-black output
    on the CPU reference path (reference scene/xml corpus analog)."""

    def _render(self, name, spp=4, w=64, h=64):
        d = parse_xml(os.path.join(ROOT, "scenes", name))
        d.camera.width, d.camera.height = w, h
        img = hippt.PythonRenderer(d, device_id=-1).render(spp=spp).numpy()
        assert np.isfinite(img).all(), name
        return d, img

    def test_grid_cbox_vpt(self):
        d, img = self._render("grid-cbox.xml")
        assert d.config.renderer == "vpt"
        assert len(d.media) == 1 and d.media[0].type == "grid"
        assert d.media[0].phase == "hg" and abs(d.media[0].g1 - 0.4) < 1e-6
        assert any(o.cullable for o in d.objects)
        assert img[..., :3].mean() > 0.01

    def test_tof_cbox_time_gate(self):
        d, img = self._render("tof-cbox.xml", spp=8)
        assert d.config.use_tof and d.config.min_time == 2.8
        assert d.cam_medium == 0
        # gate kills all light outside [2.4, 3.2]: image much darker than
        # the ungated render
        d2 = parse_xml(os.path.join(ROOT, "scenes", "tof-cbox.xml"))
        d2.camera.width = d2.camera.height = 64
        d2.config.use_tof = False
        full = hippt.PythonRenderer(d2, device_id=-1).render(spp=8).numpy()
        assert img[..., :3].mean() < full[..., :3].mean()

    def test_diamonds_dispersion(self):
        d, img = self._render("diamonds.xml", w=64, h=36)
        presets = {b.preset for b in d.bsdfs if b.type == "dispersion"}
        assert presets == {"diamond", "sapphire", "bk7"}
        assert img[..., :3].mean() > 0.005

    def test_env_balls_metals(self):
        d, img = self._render("env-balls.xml", w=80, h=45)
        assert any(e.type == "envmap" for e in d.emitters)
        env = next(e for e in d.emitters if e.type == "envmap")
        assert env.tex_id >= 0          # procedural sky texture bound
        assert abs(env.azimuth - np.radians(35.0)) < 1e-6
        metals = {b.metal for b in d.bsdfs if b.type == "ggx"}
        assert metals == {"Au", "Cu", "Ag", "W"}
        assert img[..., :3].mean() > 0.05   # sky lights everything

    def test_caustics_lt(self):
        d, img = self._render("caustics-lt.xml", spp=16)
        assert d.config.renderer == "lt"
        assert d.config.spec_constraint == 1
        assert img[..., :3].max() > 0.0  # some caustic splats landed

    def test_point_cbox(self):
        d, img = self._render("point-cbox.xml")
        assert d.emitters[0].type == "point"
        assert img[..., :3].mean() > 0.01


class TestParserRobustness:
    def test_malformed_xml(self, tmp_path):
        p = tmp_path / "bad.xml"
        p.write_text("<scene version='1.2'><renderer type='pt'>")
        import xml.etree.ElementTree as ET
        import pytest as _pt
        with _pt.raises(ET.ParseError):
            parse_xml(str(p))

    def test_missing_mesh(self, tmp_path):
        p = tmp_path / "m.xml"
        p.write_text("""<scene version='1.2'>
          <brdf type='lambertian' id='w'><rgb name='k_d' value='0.5'/></brdf>
          <shape type='obj'><string name='filename' value='nope.obj'/>
            <ref type='material' id='w'/></shape></scene>""")
        import pytest as _pt
        with _pt.raises(FileNotFoundError):
            parse_xml(str(p))

    def test_unknown_refs_raise_loudly(self, tmp_path):
        # round-2 contract change: a present-but-unknown ref id raises with
        # the known ids listed (round 1 silently bound material 0 — the
        # same silent-config-divergence class the cache_level fix closed)
        p = tmp_path / "r.xml"
        p.write_text("""<scene version='1.2'>
          <brdf type='lambertian' id='w'><rgb name='k_d' value='0.5'/></brdf>
          <shape type='sphere'><point name='center' value='0,0,0'/>
            <float name='radius' value='1'/>
            <ref type='material' id='does-not-exist'/></shape></scene>""")
        with pytest.raises(KeyError, match="does-not-exist"):
            parse_xml(str(p))


def test_dof_balls_xml():
    d, img = TestExampleScenes()._render("dof-balls.xml", spp=8, w=80, h=45)
    assert d.camera.aperture == 0.07 and d.camera.focal_dist == 5.2
    assert img[..., :3].mean() > 0.01


def test_spot_cbox_xml():
    d, img = TestExampleScenes()._render("spot-cbox.xml", spp=8)
    assert d.emitters[0].type == "area-spot"
    assert img[..., :3].mean() > 0.005


def test_multi_material_obj_hero():
    """scenes/hero.xml: one OBJ, four usemtl groups; BSDFs derive from the
    .mtl library (glass via d/Ni, metal via illum+Ks+Ns, textured
    lambertians via map_Kd) — reference tinyobjloader path scene.cu:548-660."""
    d = parse_xml(os.path.join(SCENES, "hero.xml"))
    types = [b.type for b in d.bsdfs]
    assert "ggx" in types and "translucent" in types
    # two map_Kd textures loaded and bound to diffuse slots
    mtl_bsdfs = [b for b in d.bsdfs if b.textures.get("diffuse") is not None]
    assert len([b for b in mtl_bsdfs if "diffuse" in b.textures]) >= 2
    assert len(d.textures) >= 2
    # hero groups became separate objects with normals + uvs
    hero_objs = [o for o in d.objects if o.tris is not None and o.uvs is not None]
    assert len(hero_objs) >= 4
    from hippt.scene.obj_loader import load_obj_multi
    groups, mats = load_obj_multi(os.path.join(SCENES, "meshes", "hero", "hero.obj"))
    assert [g[0] for g in groups] == ["ceramic", "metal", "glass", "wood"]
    assert set(mats) == {"ceramic", "metal", "glass", "wood"}


REF = "/root/reference/scene/xml"


@pytest.mark.skipif(not os.path.isdir(REF), reason="reference repo not present")
@pytest.mark.parametrize("name", ["cornell-box", "whiskey", "bunny", "balls",
                                  "env-balls", "medium-cbox", "point"])
def test_reference_scene_xmls_parse_and_render(name):
    """Grammar parity, proven on the reference's OWN scene files: parse
    /root/reference/scene/xml/<name>.xml (hex colors, metal presets by name,
    hflip, accelerator block, relative ../meshes paths, its shipped OBJ
    assets) and render a small frame.  (vader.xml's mesh is a
    .MISSING_LARGE_BLOBS placeholder in the reference repo itself, as are
    the meshes of the other 9 scenes — these 7 are every reference scene
    whose assets ship.)"""
    d = parse_xml(os.path.join(REF, name + ".xml"))
    assert len(d.objects) >= 5
    d.camera.width = d.camera.height = 48
    img = hippt.PythonRenderer(d, device_id=-1).render(spp=4).numpy()
    assert np.isfinite(img).all()
    assert img[..., :3].mean() > (0.001 if name == "point" else 0.01)

