"""Mitsuba-like XML v1.2 scene parser.

Capability parity: reference src/impl/scene.cu (renderer type map :977-1000,
textures :662-700, phase functions :702-775, media incl. grid refs :777-900,
<=48 BSDFs of 8 types :170-358, emitters with object binding :360-480, shapes
obj/sphere :494-660, per-object medium refs + cullable :902-930,
camera/config :1112-1114).  Element grammar as in the reference's
scene/xml/*.xml (renderer / accelerator / sensor / brdf / emitter / shape /
texture / phase / medium).
"""
from __future__ import annotations

import math
import os
import xml.etree.ElementTree as ET
from typing import Dict, List, Optional

import numpy as np

from .scene import (BsdfDesc, CameraDesc, EmitterDesc, MediumDesc, ObjectDesc,
                    RenderConfig, SceneDesc)
from .obj_loader import load_obj, load_obj_multi

RENDERER_MAP = {
    "pt": "pt", "megakernel": "pt", "pt-dynamic": "pt-dyn", "dynamic": "pt-dyn",
    "wfpt": "wfpt", "wavefront": "wfpt",
    "vpt": "vpt", "volume": "vpt",
    "lt": "lt", "light-tracing": "lt", "bdpt": "bdpt",
    "depth": "depth", "bvh-cost": "bvh-cost", "accelerator": "bvh-cost",
}

BRDF_MAP = {
    "lambertian": "lambertian",
    "specular": "specular",
    "det-refraction": "glass",
    "plastic": "plastic",
    "plastic-forward": "plastic-forward",
    "conductor-ggx": "ggx",
    "dispersion": "dispersion",
    "forward": "forward",
}


def parse_rgb(value: str):
    """'#RRGGBB' | 'r, g, b' | scalar."""
    v = value.strip()
    if v.startswith("#"):
        return tuple(int(v[i:i + 2], 16) / 255.0 for i in (1, 3, 5))
    try:
        parts = [float(x) for x in v.replace(",", " ").split()]
    except ValueError:
        # reference quirk: <rgb name="type" value="Diamond"/> carries a string
        return v
    if len(parts) == 1:
        return (parts[0],) * 3
    return tuple(parts[:3])


def _props(elem) -> Dict[str, object]:
    """Collect child <integer/float/bool/string/rgb/point/ref/transform>."""
    out: Dict[str, object] = {}
    refs: Dict[str, str] = {}
    for ch in elem:
        tag = ch.tag
        name = ch.get("name", "")
        if tag == "integer":
            out[name] = int(float(ch.get("value")))
        elif tag == "float":
            out[name] = float(ch.get("value"))
        elif tag in ("bool", "boolean"):
            out[name] = ch.get("value", "false").lower() in ("true", "1")
        elif tag == "string":
            out[name] = ch.get("value")
        elif tag == "rgb":
            out[name] = parse_rgb(ch.get("value"))
        elif tag == "point":
            # both attribute styles: x/y/z or value="x, y, z"
            if ch.get("value") is not None:
                out[name] = parse_rgb(ch.get("value"))
            else:
                out[name] = (float(ch.get("x", 0)), float(ch.get("y", 0)), float(ch.get("z", 0)))
        elif tag == "ref":
            refs[ch.get("type", "material")] = ch.get("id")
        elif tag == "transform":
            for t in ch:
                if t.tag == "lookat":
                    out["lookat_target"] = parse_rgb(t.get("target"))
                    out["lookat_origin"] = parse_rgb(t.get("origin"))
                    out["lookat_up"] = parse_rgb(t.get("up"))
        elif tag == "film":
            out.update({f"film_{k}": v for k, v in _props(ch)[0].items()})
    return out, refs


def _load_texture_image(path: str) -> Optional[np.ndarray]:
    if not os.path.exists(path):
        return None
    ext = os.path.splitext(path)[1].lower()
    if ext == ".npy":
        a = np.load(path).astype(np.float32)
        if a.ndim == 2:
            a = np.repeat(a[:, :, None], 3, axis=2)
        if a.shape[2] == 3:
            a = np.concatenate([a, np.ones_like(a[:, :, :1])], axis=2)
        return np.ascontiguousarray(a, np.float32)
    if ext == ".png":
        from ..utils.png import read_png
        return read_png(path)
    return None  # jpg etc. unsupported offline


def parse_xml(path: str) -> SceneDesc:
    tree = ET.parse(path)
    root = tree.getroot()
    assert root.tag == "scene", "not a scene file"
    version = root.get("version", "1.2")
    if version not in ("1.0", "1.1", "1.2"):
        raise ValueError(f"unsupported scene version {version}")
    base = os.path.dirname(os.path.abspath(path))
    d = SceneDesc()
    cfg = d.config

    bsdf_ids: Dict[str, int] = {}
    emitter_ids: Dict[str, int] = {}
    phase_ids: Dict[str, dict] = {}
    medium_ids: Dict[str, int] = {}
    tex_groups: Dict[str, Dict[str, int]] = {}   # texture id -> slot -> index
    hflip = False

    def load_tex(relpath: str) -> int:
        img = _load_texture_image(os.path.normpath(os.path.join(base, relpath)))
        if img is None:
            return -1
        d.textures.append(img)
        return len(d.textures) - 1

    # ---- pass 1: textures (referenced by brdfs/emitters)
    for el in root.findall("texture"):
        tid = el.get("id")
        props, _ = _props(el)
        slots: Dict[str, int] = {}
        slot_map = {"diffuse": "diffuse", "specular": "specular", "glossy": "glossy",
                    "normal": "normal", "rough1": "roughness", "roughness": "roughness",
                    "ior": "roughness", "emission": "emission"}
        for k, v in props.items():
            if k in slot_map and isinstance(v, str):
                ti = load_tex(v)
                if ti >= 0:
                    slots[slot_map[k]] = ti
        tex_groups[tid] = slots

    # ---- renderer
    rd = root.find("renderer")
    if rd is not None:
        cfg.renderer = RENDERER_MAP.get(rd.get("type", "pt"), "pt")
        props, _ = _props(rd)
        cfg.spp = props.get("sample_count", cfg.spp)
        cfg.max_depth = props.get("max_bounce", cfg.max_depth)
        cfg.max_diffuse = props.get("max_diffuse", cfg.max_depth)
        cfg.max_specular = props.get("max_specular", cfg.max_depth)
        cfg.max_transmit = props.get("max_transmit", cfg.max_depth)
        cfg.max_volume = props.get("max_volume", cfg.max_depth)
        cfg.spec_constraint = max(props.get("specular_constraint", 0), 0)
        cfg.bidirectional = bool(props.get("bidirectional", False))
        cfg.caustic_scaling = props.get("caustic_scaling", 1.0)
        cfg.radiance_clamp = props.get("radiance_clamp", 0.0)
        if "min_time" in props or "max_time" in props:
            cfg.use_tof = True
            cfg.min_time = props.get("min_time", 0.0)
            cfg.max_time = props.get("max_time", 1e9)

    # ---- accelerator
    ac = root.find("accelerator")
    if ac is not None:
        props, _ = _props(ac)
        cfg.max_leaf = props.get("max_node_num", cfg.max_leaf)
        cfg.cache_level = props.get("cache_level", cfg.cache_level)
        cfg.use_sbvh = bool(props.get("use_sbvh", False))
        cfg.ref_unsplit = bool(props.get("use_ref_unsplit", True))
        cfg.overlap_w = props.get("overlap_w", 0.0)

    # ---- sensor
    se = root.find("sensor")
    cam = d.camera
    if se is not None:
        props, refs = _props(se)
        cam.fov = props.get("fov", cam.fov)
        hflip = bool(props.get("hflip", False))
        cam.pos = props.get("lookat_origin", cam.pos)
        cam.lookat = props.get("lookat_target", cam.lookat)
        cam.up = props.get("lookat_up", cam.up)
        cam.width = props.get("film_width", cam.width)
        cam.height = props.get("film_height", cam.height)
        cam.aperture = props.get("aperture", 0.0)
        cam.focal_dist = props.get("focal_dist", 1.0)
        cam.ortho = se.get("type") == "orthographic"
        if "medium" in refs:
            pass  # resolved below after media parse (cam_medium)
        d._sensor_medium_ref = refs.get("medium")

    # ---- phases
    for el in root.findall("phase"):
        pid = el.get("id")
        props, _ = _props(el)
        phase_ids[pid] = {
            "type": el.get("type", "isotropic"),
            "g1": props.get("g", props.get("g1", 0.0)),
            "g2": props.get("g2", 0.0),
            "wmix": props.get("weight", props.get("wmix", 0.5)),
        }

    # ---- media
    for el in root.findall("medium"):
        mid = el.get("id")
        props, refs = _props(el)
        mtype = el.get("type", "homogeneous")
        ph = phase_ids.get(refs.get("phase", ""), {"type": "isotropic", "g1": 0, "g2": 0, "wmix": 0.5})
        ptype = {"isotropic": "isotropic", "hg": "hg", "duo-hg": "duo-hg",
                 "rayleigh": "rayleigh", "sggx": "sggx"}.get(ph["type"], "isotropic")
        m = MediumDesc(type="homogeneous" if mtype == "homogeneous" else "grid",
                       phase=ptype, g1=ph["g1"], g2=ph["g2"], wmix=ph["wmix"],
                       scale=props.get("scale", 1.0),
                       emission_scale=props.get("emission-scale", props.get("emission_scale", 0.0)))
        if mtype == "homogeneous":
            m.sigma_a = props.get("sigma_a", (0.1,) * 3)
            m.sigma_s = props.get("sigma_s", (1.0,) * 3)
        else:
            # grid medium: .nvdb (NanoVDB, reference vol_grid.cu:216-342 —
            # converted to dense on the host, hippt/scene/nvdb.py) and native
            # .npy grids load from disk; missing file -> procedural fallback
            dens_path = props.get("density", "")
            full = os.path.normpath(os.path.join(base, dens_path)) if dens_path else ""
            dens = None
            nvdb_bounds = None
            if full and os.path.exists(full):
                if full.endswith(".npy"):
                    dens = np.load(full).astype(np.float32)
                elif full.endswith(".nvdb"):
                    from .nvdb import read_nvdb
                    g = read_nvdb(full)[0]
                    dens = g["dense"]
                    nvdb_bounds = (g["world_min"], g["world_max"])
            if dens is None:
                from .procedural import smoke_density
                dens = smoke_density(n=96) * 12.0
            m.density = dens * 1.0
            albedo = props.get("albedo", (0.5,) * 3)
            m.sigma_s = tuple(albedo)
            m.sigma_a = tuple(1.0 - a for a in albedo)
            temp_path = props.get("emission", "")
            fullt = os.path.normpath(os.path.join(base, temp_path)) if temp_path else ""
            if fullt and os.path.exists(fullt):
                if fullt.endswith(".npy"):
                    m.temperature = np.load(fullt).astype(np.float32)
                elif fullt.endswith(".nvdb"):
                    from .nvdb import read_nvdb
                    m.temperature = read_nvdb(fullt)[0]["dense"]
            # explicit grid_lo/hi win; else .nvdb world bounds; else unit cube
            if "grid_lo" in props or nvdb_bounds is None:
                m.grid_lo = props.get("grid_lo", (0.0, 0.0, 0.0))
                m.grid_hi = props.get("grid_hi", (1.0, 1.0, 1.0))
            else:
                m.grid_lo, m.grid_hi = nvdb_bounds
        d.media.append(m)
        medium_ids[mid] = len(d.media) - 1

    # ---- brdfs
    for el in root.findall("brdf"):
        bid = el.get("id")
        btype = BRDF_MAP.get(el.get("type", "lambertian"), "lambertian")
        props, refs = _props(el)
        b = BsdfDesc(type=btype)
        if "k_d" in props:
            b.kd = props["k_d"]
        if "k_s" in props:
            b.ks = props["k_s"]
        if "k_g" in props:
            b.kg = props["k_g"]
        if btype == "glass":
            b.ior = props.get("ior", props.get("k_d", (1.5,))[0])
        if btype in ("plastic", "plastic-forward"):
            b.ior = props.get("ior", 1.5)
            b.trans_scaler = props.get("trans_scaler", 1.0)
            b.thickness = props.get("thickness", 0.0)
            if "sigma_a" in props:
                b.kg = props["sigma_a"]
        if btype == "ggx":
            b.metal = props.get("conductor", "Au")
            b.roughness_x = props.get("roughness_x", 0.1)
            b.roughness_y = props.get("roughness_y", 0.1)
        if btype == "dispersion":
            t = props.get("type", ("Diamond",))
            name = t if isinstance(t, str) else "diamond"
            b.preset = {"diamond": "diamond", "sapphire": "sapphire",
                        "silica": "fused-silica", "bk7": "bk7", "sf11": "sf11",
                        "flint": "dense-flint", "moissanite": "moissanite",
                        "ice": "water-ice"}.get(str(name).lower(), "diamond")
        # texture binding: <ref type="texture" id=.../> or same-id texture group
        tex_ref = refs.get("texture", bid)
        if tex_ref in tex_groups:
            b.textures = {k: v for k, v in tex_groups[tex_ref].items() if k != "emission"}
        d.bsdfs.append(b)
        bsdf_ids[bid] = len(d.bsdfs) - 1

    # ---- emitters
    for el in root.findall("emitter"):
        eid = el.get("id")
        etype = el.get("type", "area")
        props, refs = _props(el)
        scale = props.get("scaler", (1.0,))
        if isinstance(scale, tuple):
            scale = scale[0]
        e = EmitterDesc(type={"area": "area", "area-spot": "area-spot", "point": "point",
                              "envmap": "envmap"}.get(etype, "area"),
                        emission=props.get("emission", (1.0,) * 3),
                        scale=float(scale))
        if etype == "point":
            e.pos = props.get("center", (0.0, 0.0, 0.0))
        if etype == "area-spot":
            e.cos_max = math.cos(math.radians(props.get("angle", 30.0)))
        if etype == "envmap":
            e.azimuth = math.radians(props.get("azimuth", 0.0))
            e.zenith = math.radians(props.get("zenith", 0.0))
            e.emission = (1.0, 1.0, 1.0)
        tex_ref = refs.get("texture", eid)
        if tex_ref in tex_groups and "emission" in tex_groups[tex_ref]:
            e.tex_id = tex_groups[tex_ref]["emission"]
        elif etype == "envmap" and tex_ref in tex_groups:
            e.tex_id = next(iter(tex_groups[tex_ref].values()), -1)
        d.emitters.append(e)
        emitter_ids[eid] = len(d.emitters) - 1

    # ---- shapes

    def resolve_ref(refs, kind, table):
        """Loud ref resolution: a PRESENT-but-unknown id is a scene bug
        (silently binding material 0 was round-1-class config divergence);
        an absent ref keeps the documented default."""
        rid = refs.get(kind, "")
        if not rid:
            return None
        if rid not in table:
            raise KeyError(f"unknown {kind} ref id '{rid}' "
                           f"(known: {sorted(table)})")
        return table[rid]

    def bsdf_from_mtl(name: str, mat: dict) -> int:
        """Map a Wavefront MTL material onto the BSDF matrix (multi-material
        OBJ hero assets; reference loads materials via tinyobjloader,
        scene.cu:548-660).  Glass: d<1 with ior; metal-ish: illum>=3 with a
        strong Ks; else (textured) lambertian."""
        b = BsdfDesc()
        if mat.get("d", 1.0) < 0.99:
            b.type = "translucent"
            b.ks = mat.get("ks", (1.0, 1.0, 1.0))
            b.ior = mat.get("ni", 1.5)
        elif mat.get("illum", 2) >= 3 and max(mat.get("ks", (0,) * 3)) > 0.25:
            b.type = "ggx"
            b.kg = mat.get("ks", (0.9, 0.9, 0.9))
            ns = max(mat.get("ns", 10.0), 1.0)
            r = float(np.sqrt(2.0 / (ns + 2.0)))  # Blinn exponent -> roughness
            b.roughness_x = b.roughness_y = r
        else:
            b.type = "lambertian"
            b.kd = mat.get("kd", (0.8, 0.8, 0.8))
            if mat.get("map_kd"):
                t = load_tex(os.path.relpath(mat["map_kd"], base))
                if t >= 0:
                    b.textures["diffuse"] = t
        d.bsdfs.append(b)
        bsdf_ids[f"__mtl__{name}"] = len(d.bsdfs) - 1
        return len(d.bsdfs) - 1

    for el in root.findall("shape"):
        stype = el.get("type", "obj")
        props, refs = _props(el)
        if stype != "sphere" and props.get("use_mtl", False):
            # multi-material OBJ: one object per usemtl group, BSDFs from the
            # .mtl library; explicit <ref type="material"> is the fallback
            # for faces without a material
            fn = os.path.normpath(os.path.join(base, props.get("filename", "")))
            if not os.path.exists(fn):
                raise FileNotFoundError(f"mesh not found: {fn}")
            groups, mats = load_obj_multi(fn)
            for mname, tris, normals, uvs in groups:
                o = ObjectDesc()
                o.tris, o.normals, o.uvs = tris, normals, uvs
                key = f"__mtl__{mname}"
                if key in bsdf_ids:
                    o.bsdf = bsdf_ids[key]
                elif mname in mats:
                    o.bsdf = bsdf_from_mtl(mname, mats[mname])
                else:
                    rb = resolve_ref(refs, "material", bsdf_ids)
                    o.bsdf = 0 if rb is None else rb
                o.cullable = bool(props.get("cullable", False))
                d.objects.append(o)
            continue
        o = ObjectDesc()
        if stype == "sphere":
            c = props.get("center", (0.0, 0.0, 0.0))
            r = props.get("radius", 1.0)
            o.spheres = np.array([[c[0], c[1], c[2], r]], np.float32)
        else:
            fn = os.path.normpath(os.path.join(base, props.get("filename", "")))
            if not os.path.exists(fn):
                raise FileNotFoundError(f"mesh not found: {fn}")
            tris, normals, uvs = load_obj(fn)
            o.tris, o.normals, o.uvs = tris, normals, uvs
        rb = resolve_ref(refs, "material", bsdf_ids)
        o.bsdf = 0 if rb is None else rb
        re_ = resolve_ref(refs, "emitter", emitter_ids)
        if re_ is not None:
            o.emitter = re_
        rm = resolve_ref(refs, "medium", medium_ids)
        if rm is not None:
            o.medium_in = rm
        o.cullable = bool(props.get("cullable", False))
        d.objects.append(o)

    # sensor medium ref
    ref = getattr(d, "_sensor_medium_ref", None)
    if ref and ref in medium_ids:
        d.cam_medium = medium_ids[ref]

    d.camera.hflip = hflip
    return d
