"""Generate the in-repo example scenes (XML + OBJ meshes), parity with the
reference's scene/xml + scene/meshes/cbox set. Run from repo root:
    python scripts/make_scenes.py
"""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from hippt.scene.obj_loader import save_obj  # noqa: E402
from hippt.scene.procedural import box_mesh, quad, transform  # noqa: E402

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
MESH = os.path.join(ROOT, "scenes", "meshes", "cbox")


def cbox_meshes():
    os.makedirs(MESH, exist_ok=True)
    s = 1.0
    save_obj(f"{MESH}/floor.obj", quad((-s, 0, 0), (-s, 0, 2 * s), (s, 0, 2 * s), (s, 0, 0)))  # +y inward
    save_obj(f"{MESH}/ceiling.obj", quad((-s, 2 * s, 2 * s), (-s, 2 * s, 0), (s, 2 * s, 0), (s, 2 * s, 2 * s)))  # -y inward
    save_obj(f"{MESH}/back.obj", quad((s, 0, 2 * s), (-s, 0, 2 * s), (-s, 2 * s, 2 * s), (s, 2 * s, 2 * s)))
    save_obj(f"{MESH}/left_wall.obj", quad((-s, 0, 2 * s), (-s, 0, 0), (-s, 2 * s, 0), (-s, 2 * s, 2 * s)))
    save_obj(f"{MESH}/right_wall.obj", quad((s, 0, 0), (s, 0, 2 * s), (s, 2 * s, 2 * s), (s, 2 * s, 0)))
    save_obj(f"{MESH}/tall-box.obj", transform(box_mesh((-0.3, 0.0, -0.3), (0.3, 1.2, 0.3)),
                                               rot_y=0.3, translate=(-0.35, 0, 1.35)))
    save_obj(f"{MESH}/short-box.obj", transform(box_mesh((-0.3, 0.0, -0.3), (0.3, 0.6, 0.3)),
                                                rot_y=-0.3, translate=(0.4, 0, 0.9)))
    e = 0.4
    save_obj(f"{MESH}/light.obj", quad((-e, 2 - 1e-3, 1.0 + e), (-e, 2 - 1e-3, 1.0 - e),
                                       (e, 2 - 1e-3, 1.0 - e), (e, 2 - 1e-3, 1.0 + e)))


CORNELL_XML = """<?xml version='1.0' encoding='utf-8'?>
<scene version="1.2">
    <renderer type="pt">
        <integer name="sample_count" value="64"/>
        <integer name="max_bounce"   value="5"/>
        <integer name="max_diffuse"  value="5"/>
        <integer name="max_specular" value="5"/>
        <integer name="max_transmit" value="5"/>
    </renderer>
    <accelerator type="bvh">
        <integer name="cache_level" value="6"/>
        <integer name="max_node_num" value="4"/>
        <float name="overlap_w" value="0.6"/>
    </accelerator>
    <sensor type="perspective">
        <float name="fov" value="42.0"/>
        <transform name="toWorld">
            <lookat target="0, 1.0, 1.0" origin="0, 1.0, -2.4" up="0, 1, 0"/>
        </transform>
        <film type="film">
            <integer name="width" value="256"/>
            <integer name="height" value="256"/>
        </film>
    </sensor>
    <brdf type="lambertian" id="white"><rgb name="k_d" value="0.725, 0.71, 0.68"/></brdf>
    <brdf type="lambertian" id="red"><rgb name="k_d" value="0.63, 0.065, 0.05"/></brdf>
    <brdf type="lambertian" id="green"><rgb name="k_d" value="0.14, 0.45, 0.091"/></brdf>
    <brdf type="lambertian" id="light"><rgb name="k_d" value="0.8"/></brdf>
    <emitter type="area" id="area">
        <rgb name="emission" value="1.0, 0.85, 0.6"/>
        <rgb name="scaler" value="20.0"/>
    </emitter>
    <shape type="obj"><string name="filename" value="meshes/cbox/floor.obj"/><ref type="material" id="white"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/ceiling.obj"/><ref type="material" id="white"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/back.obj"/><ref type="material" id="white"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/left_wall.obj"/><ref type="material" id="red"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/right_wall.obj"/><ref type="material" id="green"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/tall-box.obj"/><ref type="material" id="white"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/short-box.obj"/><ref type="material" id="white"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/light.obj"/>
        <ref type="material" id="light"/><ref type="emitter" id="area"/></shape>
</scene>
"""

BALLS_XML = """<?xml version='1.0' encoding='utf-8'?>
<scene version="1.2">
    <renderer type="pt">
        <integer name="sample_count" value="64"/>
        <integer name="max_bounce" value="8"/>
        <integer name="max_diffuse" value="4"/>
        <integer name="max_specular" value="8"/>
        <integer name="max_transmit" value="8"/>
    </renderer>
    <sensor type="perspective">
        <float name="fov" value="45"/>
        <transform name="toWorld">
            <lookat target="0, 0.5, 1" origin="0, 1.2, -3.2" up="0, 1, 0"/>
        </transform>
        <film type="film"><integer name="width" value="320"/><integer name="height" value="180"/></film>
    </sensor>
    <brdf type="lambertian" id="floor"><rgb name="k_d" value="0.7"/></brdf>
    <brdf type="conductor-ggx" id="gold">
        <string name="conductor" value="Au"/>
        <float name="roughness_x" value="0.2"/><float name="roughness_y" value="0.05"/>
        <rgb name="k_g" value="1"/>
    </brdf>
    <brdf type="det-refraction" id="glass"><rgb name="k_d" value="1.5"/><rgb name="k_s" value="0.99"/></brdf>
    <brdf type="dispersion" id="diamond"><rgb name="type" value="Diamond"/><rgb name="k_s" value="0.99"/></brdf>
    <brdf type="plastic" id="red-plastic">
        <rgb name="k_d" value="#DD3322"/><rgb name="k_s" value="1.0"/>
        <float name="ior" value="1.5"/><float name="trans_scaler" value="1"/>
    </brdf>
    <brdf type="specular" id="mirror"><rgb name="k_s" value="0.95"/></brdf>
    <emitter type="area" id="lamp">
        <rgb name="emission" value="1, 0.95, 0.85"/><rgb name="scaler" value="30"/>
    </emitter>
    <emitter type="point" id="fill">
        <rgb name="emission" value="0.6, 0.7, 1.0"/><rgb name="scaler" value="2.5"/>
        <point name="center" x="-2" y="3" z="-2"/>
    </emitter>
    <shape type="obj"><string name="filename" value="meshes/cbox/floor_big.obj"/><ref type="material" id="floor"/></shape>
    <shape type="sphere"><point name="center" x="-1.2" y="0.5" z="1.0"/><float name="radius" value="0.5"/><ref type="material" id="gold"/></shape>
    <shape type="sphere"><point name="center" x="0.0" y="0.5" z="1.0"/><float name="radius" value="0.5"/><ref type="material" id="glass"/></shape>
    <shape type="sphere"><point name="center" x="1.2" y="0.5" z="1.0"/><float name="radius" value="0.5"/><ref type="material" id="red-plastic"/></shape>
    <shape type="sphere"><point name="center" x="-0.6" y="0.35" z="-0.2"/><float name="radius" value="0.35"/><ref type="material" id="mirror"/></shape>
    <shape type="sphere"><point name="center" x="0.6" y="0.35" z="-0.2"/><float name="radius" value="0.35"/><ref type="material" id="diamond"/></shape>
    <shape type="obj"><string name="filename" value="meshes/cbox/lamp.obj"/>
        <ref type="material" id="floor"/><ref type="emitter" id="lamp"/></shape>
</scene>
"""


def sky_texture():
    """Procedural lat-long sky for env-balls.xml (no HDR assets ship)."""
    from hippt.utils.png import write_png
    h, w = 128, 256
    y = np.linspace(1, -1, h)[:, None]
    x = np.linspace(0, 2 * np.pi, w)[None, :]
    sky = np.zeros((h, w, 3), np.float32)
    sky[..., 2] = 0.55 + 0.4 * np.clip(y, 0, 1)                       # blue up
    sky[..., 1] = 0.45 + 0.25 * np.clip(y, 0, 1) + 0.2 * np.clip(-y, 0, 1)
    sky[..., 0] = 0.4 + 0.5 * np.clip(-y, 0, 1)                       # warm down
    # sun disk
    sun = np.exp(-(((x - 1.1) ** 2) / 0.01 + ((y - 0.45) ** 2) / 0.004))
    sky += 8.0 * sun[..., None] * np.array([1.0, 0.95, 0.85], np.float32)
    img = (np.clip(sky / (1 + sky), 0, 1) * 255).astype(np.uint8)
    tdir = os.path.join(ROOT, "scenes", "textures")
    os.makedirs(tdir, exist_ok=True)
    write_png(os.path.join(tdir, "sky.png"), img)


def smoke_nvdb():
    """NanoVDB smoke asset for grid-cbox-nvdb.xml (hippt/scene/nvdb.py)."""
    from hippt.scene.procedural import smoke_density
    from hippt.scene.nvdb import write_nvdb
    adir = os.path.join(ROOT, "scenes", "assets")
    os.makedirs(adir, exist_ok=True)
    d = (smoke_density(n=64) * 12.0).astype(np.float32)
    write_nvdb(os.path.join(adir, "smoke.nvdb"), d, voxel_size=1.2 / 64,
               world_origin=(-0.6, 0.05, 0.4), grid_name="density")


def main():
    cbox_meshes()
    smoke_nvdb()
    save_obj(f"{MESH}/floor_big.obj", quad((-6, 0, -6), (-6, 0, 6), (6, 0, 6), (6, 0, -6)))
    save_obj(f"{MESH}/lamp.obj", quad((1.5, 4, -1.5), (1.5, 4, 1.5), (-1.5, 4, 1.5), (-1.5, 4, -1.5)))
    save_obj(f"{MESH}/smoke_bound.obj", box_mesh((-0.6, 0.05, 0.4), (0.6, 1.6, 1.6)))
    save_obj(f"{MESH}/backdrop.obj", quad((-5, 0, 2.5), (5, 0, 2.5), (5, 5, 2.5), (-5, 5, 2.5)))
    sky_texture()
    with open(os.path.join(ROOT, "scenes", "cornell-box.xml"), "w") as f:
        f.write(CORNELL_XML)
    with open(os.path.join(ROOT, "scenes", "balls.xml"), "w") as f:
        f.write(BALLS_XML)
    print("scenes written")


if __name__ == "__main__":
    main()
