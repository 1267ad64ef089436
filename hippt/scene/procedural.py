"""Procedural scene generators.

BASELINE.json requires benchmarks "on procedurally generated scenes of the
named shape with random materials" (no large mesh assets ship with either
repo).  Generators: cornell box (test/golden scenes), modern-kitchen-class
(~100-200k triangles, mixed BSDFs — the headline wavefront/megakernel bench),
sports-car-class (high-poly curved shells for the 4K DDP bench), and a
procedural smoke grid for the volumetric bench.
"""
from __future__ import annotations

import math
from typing import Optional

import numpy as np

from .scene import (BsdfDesc, CameraDesc, EmitterDesc, MediumDesc, ObjectDesc,
                    RenderConfig, SceneDesc)


# ------------------------------------------------------------ mesh helpers

def quad(p0, p1, p2, p3):
    """Two triangles for quad p0p1p2p3 (ccw)."""
    p0, p1, p2, p3 = (np.asarray(p, np.float32) for p in (p0, p1, p2, p3))
    return np.stack([np.stack([p0, p1, p2]), np.stack([p0, p2, p3])])


def box_mesh(lo, hi, inward=False):
    lo = np.asarray(lo, np.float32)
    hi = np.asarray(hi, np.float32)
    x0, y0, z0 = lo
    x1, y1, z1 = hi
    faces = [
        quad((x0, y0, z0), (x1, y0, z0), (x1, y1, z0), (x0, y1, z0)),  # z0 (back)
        quad((x1, y0, z1), (x0, y0, z1), (x0, y1, z1), (x1, y1, z1)),  # z1
        quad((x0, y0, z1), (x0, y0, z0), (x0, y1, z0), (x0, y1, z1)),  # x0
        quad((x1, y0, z0), (x1, y0, z1), (x1, y1, z1), (x1, y1, z0)),  # x1
        quad((x0, y0, z1), (x1, y0, z1), (x1, y0, z0), (x0, y0, z0)),  # y0 (floor)
        quad((x0, y1, z0), (x1, y1, z0), (x1, y1, z1), (x0, y1, z1)),  # y1 (ceiling)
    ]
    tris = np.concatenate(faces)
    # the face quads above wind with normals pointing INTO the box; flip by
    # default so solid objects have outward geometric normals (medium
    # enter/exit tests depend on this), keep them for room interiors
    if not inward:
        tris = tris[:, ::-1, :].copy()
    return tris


def transform(tris, scale=1.0, rot_y=0.0, translate=(0, 0, 0)):
    t = np.asarray(tris, np.float32) * scale
    if rot_y:
        c, s = math.cos(rot_y), math.sin(rot_y)
        R = np.array([[c, 0, s], [0, 1, 0], [-s, 0, c]], np.float32)
        t = t @ R.T
    return t + np.asarray(translate, np.float32)


def uv_sphere_mesh(center, radius, n_theta=16, n_phi=32):
    ct = np.asarray(center, np.float32)
    th = np.linspace(0, math.pi, n_theta + 1)
    ph = np.linspace(0, 2 * math.pi, n_phi + 1)
    tris = []
    for i in range(n_theta):
        for j in range(n_phi):
            p = []
            for a, b in ((i, j), (i + 1, j), (i + 1, j + 1), (i, j + 1)):
                p.append(ct + radius * np.array([
                    math.sin(th[a]) * math.cos(ph[b]),
                    math.cos(th[a]),
                    math.sin(th[a]) * math.sin(ph[b])], np.float32))
            if i > 0:
                tris.append([p[0], p[1], p[2]])
            if i < n_theta - 1:
                tris.append([p[0], p[2], p[3]])
    return np.asarray(tris, np.float32)


def displaced_grid_mesh(nx, nz, extent, height_fn, y0=0.0):
    """Heightfield sheet: (nx*nz*2) triangles."""
    xs = np.linspace(-extent / 2, extent / 2, nx + 1)
    zs = np.linspace(-extent / 2, extent / 2, nz + 1)
    X, Z = np.meshgrid(xs, zs, indexing="ij")
    Y = y0 + height_fn(X, Z)
    P = np.stack([X, Y, Z], axis=-1).astype(np.float32)
    a = P[:-1, :-1]
    b = P[1:, :-1]
    c = P[1:, 1:]
    d = P[:-1, 1:]
    t1 = np.stack([a, b, c], axis=2).reshape(-1, 3, 3)
    t2 = np.stack([a, c, d], axis=2).reshape(-1, 3, 3)
    return np.concatenate([t1, t2])


# ------------------------------------------------------------ cornell box

def cornell_box(width=256, height=256, spp=4, max_depth=2, renderer="pt",
                light_scale=20.0, use_sbvh=False) -> SceneDesc:
    """The classic box: white floor/ceiling/back, red/green walls, two blocks,
    area light on the ceiling.  BASELINE config #1/#2 scene."""
    d = SceneDesc()
    d.bsdfs = [
        BsdfDesc(type="lambertian", kd=(0.725, 0.71, 0.68)),   # white
        BsdfDesc(type="lambertian", kd=(0.63, 0.065, 0.05)),   # red
        BsdfDesc(type="lambertian", kd=(0.14, 0.45, 0.091)),   # green
        BsdfDesc(type="lambertian", kd=(0.8, 0.8, 0.8)),       # light surface
    ]
    d.emitters = [EmitterDesc(type="area", emission=(1.0, 0.85, 0.6), scale=light_scale)]
    s = 1.0
    # room (open toward camera at z=-2s..? camera looks +z); box [-1,1]^2 x [0,2]
    floor = quad((-s, 0, 0), (-s, 0, 2 * s), (s, 0, 2 * s), (s, 0, 0))      # +y inward
    ceil = quad((-s, 2 * s, 2 * s), (-s, 2 * s, 0), (s, 2 * s, 0), (s, 2 * s, 2 * s))  # -y inward
    back = quad((s, 0, 2 * s), (-s, 0, 2 * s), (-s, 2 * s, 2 * s), (s, 2 * s, 2 * s))
    left = quad((-s, 0, 2 * s), (-s, 0, 0), (-s, 2 * s, 0), (-s, 2 * s, 2 * s))
    right = quad((s, 0, 0), (s, 0, 2 * s), (s, 2 * s, 2 * s), (s, 2 * s, 0))
    white = np.concatenate([floor, ceil, back])
    tall = transform(box_mesh((-0.3, 0.0, -0.3), (0.3, 1.2, 0.3)), rot_y=0.3,
                     translate=(-0.35, 0, 1.35))
    short = transform(box_mesh((-0.3, 0.0, -0.3), (0.3, 0.6, 0.3)), rot_y=-0.3,
                      translate=(0.4, 0, 0.9))
    e = 0.4
    # winding chosen so the emitter normal faces DOWN (-y): emitters are
    # single-sided (emitter_eval_le gates on cos > 0)
    light = quad((-e, 2 * s - 1e-3, 1.0 + e), (-e, 2 * s - 1e-3, 1.0 - e),
                 (e, 2 * s - 1e-3, 1.0 - e), (e, 2 * s - 1e-3, 1.0 + e))
    d.objects = [
        ObjectDesc(tris=np.concatenate([white, tall, short]), bsdf=0),
        ObjectDesc(tris=left, bsdf=1),
        ObjectDesc(tris=right, bsdf=2),
        ObjectDesc(tris=light, bsdf=3, emitter=0),
    ]
    d.camera = CameraDesc(pos=(0, 1.0, -2.4), lookat=(0, 1.0, 1.0), up=(0, 1, 0),
                          fov=42.0, width=width, height=height)
    d.config = RenderConfig(spp=spp, max_depth=max_depth, max_diffuse=max_depth,
                            max_specular=max_depth, max_transmit=max_depth,
                            renderer=renderer, use_sbvh=use_sbvh)
    return d


# ------------------------------------------------------ modern-kitchen-class

def kitchen(width=1920, height=1080, spp=64, renderer="wfpt", seed=5,
            detail=1.0) -> SceneDesc:
    """Procedural modern-kitchen-class interior: a room with cabinets,
    a counter with displaced micro-geometry, appliances (GGX metals), glass
    objects, plastic chairs, ~150k tris with mixed materials.  Headline
    benchmark scene shape (BASELINE config #3)."""
    rng = np.random.default_rng(seed)
    d = SceneDesc()
    d.bsdfs = [
        BsdfDesc(type="lambertian", kd=(0.75, 0.73, 0.70)),             # 0 walls
        BsdfDesc(type="lambertian", kd=(0.35, 0.24, 0.16)),             # 1 wood floor
        BsdfDesc(type="plastic", kd=(0.9, 0.9, 0.92), ior=1.5, trans_scaler=1.0),  # 2 cabinet
        BsdfDesc(type="ggx", metal="Ag", roughness_x=0.15, roughness_y=0.15),      # 3 steel
        BsdfDesc(type="ggx", metal="Al", roughness_x=0.35, roughness_y=0.35),      # 4 brushed
        BsdfDesc(type="glass", ks=(0.98, 0.98, 0.98), ior=1.5),         # 5 glass
        BsdfDesc(type="plastic", kd=(0.85, 0.25, 0.2), ior=1.45),       # 6 red plastic
        BsdfDesc(type="specular", ks=(0.9, 0.9, 0.9)),                  # 7 mirror
        BsdfDesc(type="lambertian", kd=(0.9, 0.9, 0.9)),                # 8 light surf
        BsdfDesc(type="ggx", metal="Au", roughness_x=0.25, roughness_y=0.08),      # 9 aniso gold
    ]
    d.emitters = [
        EmitterDesc(type="area", emission=(1.0, 0.92, 0.8), scale=22.0),
        EmitterDesc(type="area", emission=(0.7, 0.8, 1.0), scale=14.0),
    ]
    W, H, D = 6.0, 3.0, 5.0
    room = box_mesh((0, 0, 0), (W, H, D), inward=True)
    objs = [ObjectDesc(tris=room, bsdf=0)]
    # floor sheet
    floor = quad((0, 1e-3, 0), (0, 1e-3, D), (W, 1e-3, D), (W, 1e-3, 0))
    objs.append(ObjectDesc(tris=floor, bsdf=1))
    # cabinets along the back wall
    cab = []
    x = 0.2
    while x < W - 0.8:
        w = rng.uniform(0.5, 0.9)
        cab.append(box_mesh((x, 0, D - 0.65), (x + w, 0.9, D - 0.05)))
        if rng.random() < 0.7:
            cab.append(box_mesh((x, 1.6, D - 0.4), (x + w, 2.3, D - 0.05)))
        x += w + 0.05
    objs.append(ObjectDesc(tris=np.concatenate(cab), bsdf=2))
    # countertop with displaced surface detail (the triangle-count driver)
    n_grid = int(172 * math.sqrt(detail))
    counter = displaced_grid_mesh(
        n_grid, n_grid, 2.4,
        lambda X, Z: 0.02 * np.sin(X * 21.0) * np.cos(Z * 17.0) +
                     0.01 * np.sin(X * 53.0 + Z * 31.0),
        y0=0.92)
    counter = transform(counter, translate=(W / 2, 0, D - 1.5))
    objs.append(ObjectDesc(tris=counter, bsdf=9))
    # tiled backsplash wall with relief (second detail surface)
    n_bs = int(150 * math.sqrt(detail))
    splash = displaced_grid_mesh(
        n_bs, n_bs, 2.8,
        lambda X, Z: 0.012 * np.sign(np.sin(X * 12.0) * np.sin(Z * 12.0)) +
                     0.004 * np.sin(X * 40.0),
        y0=0.0)
    # rotate the heightfield sheet vertical against the back wall
    splash = splash[:, :, [0, 2, 1]]   # swap y/z -> vertical panel
    splash = transform(splash, translate=(W / 2, 1.9, D - 0.06))
    objs.append(ObjectDesc(tris=splash, bsdf=2))
    # island
    objs.append(ObjectDesc(tris=box_mesh((W / 2 - 1.1, 0, 1.6), (W / 2 + 1.1, 0.95, 2.8)), bsdf=2))
    # appliances: fridge + oven (metal)
    metal = [box_mesh((W - 1.0, 0, D - 0.8), (W - 0.15, 2.0, D - 0.1)),
             box_mesh((0.3, 0.0, D - 0.75), (1.1, 0.85, D - 0.08))]
    objs.append(ObjectDesc(tris=np.concatenate(metal), bsdf=3))
    # glassware + bowls on the counter/island (spheres & sphere meshes)
    glass_tris = []
    steel_tris = []
    for i in range(int(40 * detail)):
        cx = rng.uniform(W / 2 - 1.0, W / 2 + 1.0)
        cz = rng.uniform(1.7, 2.7)
        r = rng.uniform(0.04, 0.10)
        m = uv_sphere_mesh((cx, 0.95 + r, cz), r,
                           n_theta=int(10 * math.sqrt(detail)) + 4,
                           n_phi=int(20 * math.sqrt(detail)) + 6)
        (glass_tris if rng.random() < 0.5 else steel_tris).append(m)
    if glass_tris:
        objs.append(ObjectDesc(tris=np.concatenate(glass_tris), bsdf=5))
    if steel_tris:
        objs.append(ObjectDesc(tris=np.concatenate(steel_tris), bsdf=4))
    # chairs (plastic)
    chairs = []
    for i in range(4):
        cx = 1.0 + i * 1.2
        chairs.append(transform(box_mesh((-0.2, 0.0, -0.2), (0.2, 0.45, 0.2)),
                                rot_y=rng.uniform(0, 6.28), translate=(cx, 0, 1.0)))
    objs.append(ObjectDesc(tris=np.concatenate(chairs), bsdf=6))
    # mirror panel
    objs.append(ObjectDesc(tris=quad((0.05, 1.0, 1.0), (0.05, 1.0, 3.0),
                                     (0.05, 2.2, 3.0), (0.05, 2.2, 1.0)), bsdf=7))
    # ceiling lights
    l1 = quad((W / 2 + 0.8, H - 1e-3, 1.8), (W / 2 + 0.8, H - 1e-3, 2.6),
              (W / 2 - 0.8, H - 1e-3, 2.6), (W / 2 - 0.8, H - 1e-3, 1.8))
    l2 = quad((W / 2 + 0.5, H - 1e-3, 3.8), (W / 2 + 0.5, H - 1e-3, 4.3),
              (W / 2 - 0.5, H - 1e-3, 4.3), (W / 2 - 0.5, H - 1e-3, 3.8))
    objs.append(ObjectDesc(tris=l1, bsdf=8, emitter=0))
    objs.append(ObjectDesc(tris=l2, bsdf=8, emitter=1))
    d.objects = objs
    d.camera = CameraDesc(pos=(W / 2 + 0.3, 1.55, 0.35), lookat=(W / 2 - 0.4, 1.1, D - 1.2),
                          up=(0, 1, 0), fov=58.0, width=width, height=height)
    d.config = RenderConfig(spp=spp, max_depth=10, max_diffuse=6, max_specular=10,
                            max_transmit=10, renderer=renderer)
    return d


# ---------------------------------------------------------- sports-car-class

def sports_car(width=3840, height=2160, spp=64, renderer="pt", seed=7) -> SceneDesc:
    """vision-gt sports-car-class scene: a curved high-poly car shell (NURBS-ish
    displaced patches, ~300k tris), studio floor and area-light rig.
    BASELINE config #5 scene shape (DDP sample-split bench)."""
    rng = np.random.default_rng(seed)
    d = SceneDesc()
    d.bsdfs = [
        BsdfDesc(type="lambertian", kd=(0.55, 0.55, 0.58)),               # 0 studio
        BsdfDesc(type="ggx", metal="Al", roughness_x=0.08, roughness_y=0.08,
                 kg=(0.9, 0.05, 0.08)),                                   # 1 car paint
        BsdfDesc(type="glass", ks=(0.95, 0.97, 0.98), ior=1.52),          # 2 windows
        BsdfDesc(type="ggx", metal="Cr", roughness_x=0.05, roughness_y=0.05),  # 3 chrome
        BsdfDesc(type="lambertian", kd=(0.04, 0.04, 0.045)),              # 4 tires
        BsdfDesc(type="lambertian", kd=(0.9, 0.9, 0.9)),                  # 5 light
        BsdfDesc(type="plastic", kd=(0.1, 0.1, 0.12), ior=1.45),          # 6 trim
    ]
    d.emitters = [EmitterDesc(type="area", emission=(1.0, 0.98, 0.95), scale=14.0)]
    objs = []
    # studio: floor + cyc wall
    objs.append(ObjectDesc(tris=np.concatenate([
        quad((-8, 0, -8), (8, 0, -8), (8, 0, 8), (-8, 0, 8)),
        quad((-8, 0, 8), (8, 0, 8), (8, 6, 8), (-8, 6, 8)),
        quad((-8, 0, -8), (-8, 0, 8), (-8, 6, 8), (-8, 6, -8)),
    ]), bsdf=0))

    # car body: displaced superellipsoid shell
    nu, nv = 220, 130
    u = np.linspace(-math.pi / 2, math.pi / 2, nu + 1)
    v = np.linspace(-math.pi, math.pi, nv + 1)
    U, V = np.meshgrid(u, v, indexing="ij")

    def sgn_pow(x, p):
        return np.sign(x) * np.abs(x) ** p

    a, b, c = 2.3, 0.62, 1.0   # length, height, width
    e1, e2 = 0.5, 0.85
    # superellipsoid: length along sin(U) (U covers the FULL -a..a span),
    # cross-section radius along cos(U)
    X = a * sgn_pow(np.sin(U), e1)
    cr = sgn_pow(np.cos(U), e1)
    Yr = b * cr * sgn_pow(np.cos(V), e2)
    Z = c * cr * sgn_pow(np.sin(V), e2)
    # cabin bulge on the upper shell, amidships
    bulge = 0.5 * np.exp(-((X + 0.3) ** 2) / 0.8) * np.clip(np.cos(V), 0, 1) * cr
    Y = np.abs(Yr) * 0.55 + bulge + 0.30
    P = np.stack([X, Y, Z], axis=-1).astype(np.float32)
    A = P[:-1, :-1]; B = P[1:, :-1]; Cc = P[1:, 1:]; Dd = P[:-1, 1:]
    body = np.concatenate([
        np.stack([A, B, Cc], axis=2).reshape(-1, 3, 3),
        np.stack([A, Cc, Dd], axis=2).reshape(-1, 3, 3)])
    objs.append(ObjectDesc(tris=body, bsdf=1))
    # windows: a second shell band
    win = transform(body[::7] * np.array([0.82, 1.0, 0.86], np.float32),
                    translate=(0.0, 0.08, 0))
    objs.append(ObjectDesc(tris=win, bsdf=2))
    # wheels: torus-ish from uv spheres squashed
    wheels = []
    chrome = []
    for sx in (-1.35, 1.35):
        for sz in (-1.05, 1.05):
            m = uv_sphere_mesh((sx, 0.35, sz), 0.35, n_theta=18, n_phi=36)
            m[:, :, 2] = (m[:, :, 2] - sz) * 0.35 + sz
            wheels.append(m)
            h = uv_sphere_mesh((sx, 0.35, sz * 1.12), 0.16, n_theta=10, n_phi=20)
            h[:, :, 2] = (h[:, :, 2] - sz * 1.12) * 0.3 + sz * 1.12
            chrome.append(h)
    objs.append(ObjectDesc(tris=np.concatenate(wheels), bsdf=4))
    objs.append(ObjectDesc(tris=np.concatenate(chrome), bsdf=3))
    # trim details: random small boxes along the body
    trims = []
    for i in range(60):
        t = rng.uniform(-1, 1)
        trims.append(transform(box_mesh((-0.05, -0.02, -0.05), (0.05, 0.02, 0.05)),
                               rot_y=rng.uniform(0, 6.28),
                               translate=(2.2 * t, 0.55 + 0.2 * abs(math.sin(4 * t)),
                                          1.0 * math.copysign(1, rng.random() - 0.5))))
    objs.append(ObjectDesc(tris=np.concatenate(trims), bsdf=6))
    # softbox light rig
    l = quad((2.5, 4.2, -2.0), (2.5, 4.2, 2.0), (-2.5, 4.2, 2.0), (-2.5, 4.2, -2.0))
    objs.append(ObjectDesc(tris=l, bsdf=5, emitter=0))
    d.objects = objs
    d.camera = CameraDesc(pos=(4.6, 1.6, -4.4), lookat=(0.0, 0.55, 0.0), up=(0, 1, 0),
                          fov=40.0, width=width, height=height)
    d.config = RenderConfig(spp=spp, max_depth=10, max_diffuse=5, max_specular=10,
                            max_transmit=8, renderer=renderer)
    return d


# ------------------------------------------------------------- smoke volume

def smoke_density(n=96, seed=3):
    """Procedural smoke plume density grid (nz,ny,nx) float32 in [0,~1]."""
    rng = np.random.default_rng(seed)
    # value-noise octaves
    def noise3(shape, cells):
        g = rng.standard_normal((cells + 1,) * 3).astype(np.float32)
        zs = np.linspace(0, cells, shape[0])
        ys = np.linspace(0, cells, shape[1])
        xs = np.linspace(0, cells, shape[2])
        iz, iy, ix = np.floor(zs).astype(int), np.floor(ys).astype(int), np.floor(xs).astype(int)
        iz = np.minimum(iz, cells - 1); iy = np.minimum(iy, cells - 1); ix = np.minimum(ix, cells - 1)
        fz = (zs - iz)[:, None, None]; fy = (ys - iy)[None, :, None]; fx = (xs - ix)[None, None, :]
        def g3(dz, dy, dx):
            return g[np.ix_(iz + dz, iy + dy, ix + dx)]
        c000, c001 = g3(0, 0, 0), g3(0, 0, 1)
        c010, c011 = g3(0, 1, 0), g3(0, 1, 1)
        c100, c101 = g3(1, 0, 0), g3(1, 0, 1)
        c110, c111 = g3(1, 1, 0), g3(1, 1, 1)
        def lerp(a, b, t):
            return a + (b - a) * t
        return lerp(lerp(lerp(c000, c001, fx), lerp(c010, c011, fx), fy),
                    lerp(lerp(c100, c101, fx), lerp(c110, c111, fx), fy), fz)

    shape = (n, n, n)
    dens = np.zeros(shape, np.float32)
    amp, cells = 1.0, 4
    for o in range(4):
        dens += amp * noise3(shape, cells)
        amp *= 0.5
        cells *= 2
    dens = np.abs(dens)
    # plume envelope: rising column widening with height
    z, y, x = np.meshgrid(np.linspace(-1, 1, n), np.linspace(0, 1, n),
                          np.linspace(-1, 1, n), indexing="ij")
    r = np.sqrt(x ** 2 + z ** 2)
    envelope = np.clip(1.0 - r / (0.25 + 0.75 * y), 0, 1) * np.clip(1.2 - y, 0, 1)
    dens = dens * envelope.transpose(1, 0, 2)
    dens = np.clip(dens - 0.05, 0, None)
    m = dens.max()
    return (dens / m if m > 0 else dens).astype(np.float32)


def smoke_box(width=1280, height=720, spp=32, n_grid=96, emission=False,
              renderer="vpt") -> SceneDesc:
    """NanoVDB-smoke-class volumetric scene: a smoke plume grid in a lit box
    (BASELINE config #4 shape)."""
    d = cornell_box(width=width, height=height, spp=spp, max_depth=10,
                    renderer=renderer, light_scale=30.0)
    d.camera.width, d.camera.height = width, height
    dens = smoke_density(n=n_grid)
    temp = None
    escale = 0.0
    if emission:
        temp = (dens * 4500.0).astype(np.float32)
        escale = 2.0
    d.media = [MediumDesc(type="grid", sigma_a=(0.4, 0.45, 0.5), sigma_s=(3.5, 3.5, 3.5),
                          phase="hg", g1=0.3, density=dens * 18.0,
                          temperature=temp, emission_scale=escale, temp_scale=1.0,
                          grid_lo=(-0.7, 0.05, 0.5), grid_hi=(0.7, 1.75, 1.9))]
    # container object: forward-bsdf box around the grid (cullable boundary)
    nb = len(d.bsdfs)
    d.bsdfs.append(BsdfDesc(type="forward"))
    d.objects.append(ObjectDesc(tris=box_mesh((-0.7, 0.05, 0.5), (0.7, 1.75, 1.9)),
                                bsdf=nb, medium_in=0, cullable=True))
    d.config.renderer = renderer
    d.config.max_depth = 32
    d.config.max_volume = 64
    return d
