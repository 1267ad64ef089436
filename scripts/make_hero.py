"""Generate the multi-material OBJ hero asset (scenes/meshes/hero/hero.obj
+ hero.mtl + textures) and the scene that renders it (scenes/hero.xml).

Parity: the reference ships multi-material hero meshes in-repo
(scene/meshes vader/whiskey, loaded through tinyobjloader with per-face
materials).  This repo's assets are procedural (no network, no large
blobs): a lathe-profile ceramic vase with a checker map_Kd, a polished
metal torus knot, a glass orb, and a wood-textured pedestal — one OBJ,
four `usemtl` groups, smooth vertex normals and UVs throughout.
"""
import os
import sys

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
HERO = os.path.join(ROOT, "scenes", "meshes", "hero")


# ---------------------------------------------------------------- mesh math
def grid_mesh(P, N, UV, close_u=False):
    """Index a (nu, nv, 3) vertex grid into triangles with per-vertex
    normals/uvs. Returns flat (v, vn, vt, faces) lists for OBJ emission."""
    nu, nv = P.shape[:2]
    faces = []
    for i in range(nu - 1 + (1 if close_u else 0)):
        i1 = (i + 1) % nu
        for j in range(nv - 1):
            a = (i, j); b = (i1, j); c = (i1, j + 1); d = (i, j + 1)
            faces.append((a, b, c))
            faces.append((a, c, d))
    return P, N, UV, faces


def emit_group(fobj, P, N, UV, faces, voff, toff, noff):
    nu, nv = P.shape[:2]
    for i in range(nu):
        for j in range(nv):
            p = P[i, j]; fobj.write(f"v {p[0]:.6f} {p[1]:.6f} {p[2]:.6f}\n")
    for i in range(nu):
        for j in range(nv):
            t = UV[i, j]; fobj.write(f"vt {t[0]:.6f} {t[1]:.6f}\n")
    for i in range(nu):
        for j in range(nv):
            n = N[i, j]; fobj.write(f"vn {n[0]:.6f} {n[1]:.6f} {n[2]:.6f}\n")

    def vid(ij):
        return ij[0] * nv + ij[1] + 1

    for f in faces:
        toks = " ".join(f"{vid(ij)+voff}/{vid(ij)+toff}/{vid(ij)+noff}" for ij in f)
        fobj.write(f"f {toks}\n")
    n = nu * nv
    return voff + n, toff + n, noff + n


def lathe(profile, n_seg=64, center=(0, 0, 0), uv_scale=(1.0, 1.0)):
    """Revolve a 2D (r, y) profile around +Y. Smooth normals from the
    profile tangent."""
    prof = np.asarray(profile, np.float64)
    m = len(prof)
    th = np.linspace(0, 2 * np.pi, n_seg, endpoint=False)
    P = np.zeros((n_seg, m, 3))
    N = np.zeros((n_seg, m, 3))
    UV = np.zeros((n_seg, m, 2))
    # profile tangents -> 2D normals (pointing outward)
    t2 = np.gradient(prof, axis=0)
    n2 = np.stack([t2[:, 1], -t2[:, 0]], axis=1)
    n2 /= np.maximum(np.linalg.norm(n2, axis=1, keepdims=True), 1e-12)
    for i, a in enumerate(th):
        ca, sa = np.cos(a), np.sin(a)
        P[i, :, 0] = prof[:, 0] * ca + center[0]
        P[i, :, 1] = prof[:, 1] + center[1]
        P[i, :, 2] = prof[:, 0] * sa + center[2]
        N[i, :, 0] = n2[:, 0] * ca
        N[i, :, 1] = n2[:, 1]
        N[i, :, 2] = n2[:, 0] * sa
        UV[i, :, 0] = (a / (2 * np.pi)) * uv_scale[0]
        UV[i, :, 1] = np.linspace(0, 1, m) * uv_scale[1]
    return grid_mesh(P, N, UV, close_u=True)


def torus_knot(p=2, q=3, R=0.22, r=0.055, n_u=160, n_v=20, center=(0, 0, 0)):
    """(p,q) torus knot tube with frames from the curve tangent."""
    t = np.linspace(0, 2 * np.pi, n_u, endpoint=False)
    cx = (R + 0.5 * R * np.cos(q * t)) * np.cos(p * t)
    cz = (R + 0.5 * R * np.cos(q * t)) * np.sin(p * t)
    cy = 0.5 * R * np.sin(q * t)
    C = np.stack([cx, cy, cz], axis=1)
    T = np.gradient(C, axis=0)
    T /= np.maximum(np.linalg.norm(T, axis=1, keepdims=True), 1e-12)
    up = np.array([0.0, 1.0, 0.0])
    B = np.cross(T, up)
    B /= np.maximum(np.linalg.norm(B, axis=1, keepdims=True), 1e-12)
    Nf = np.cross(B, T)
    ph = np.linspace(0, 2 * np.pi, n_v, endpoint=False)
    P = np.zeros((n_u, n_v, 3))
    N = np.zeros((n_u, n_v, 3))
    UV = np.zeros((n_u, n_v, 2))
    for j, a in enumerate(ph):
        nrm = Nf * np.cos(a) + B * np.sin(a)
        P[:, j] = C + r * nrm + np.asarray(center)
        N[:, j] = nrm
        UV[:, j, 0] = t / (2 * np.pi)
        UV[:, j, 1] = a / (2 * np.pi)
    # close the tube seam in v by duplicating ring 0 (grid_mesh closes u only)
    P = np.concatenate([P, P[:, :1]], axis=1)
    N = np.concatenate([N, N[:, :1]], axis=1)
    UV2 = np.concatenate([UV, UV[:, :1]], axis=1)
    UV2[:, -1, 1] = 1.0
    return grid_mesh(P, N, UV2, close_u=True)


def uv_sphere(r=0.16, center=(0, 0, 0), n_u=48, n_v=32):
    th = np.linspace(0, 2 * np.pi, n_u, endpoint=False)
    phi = np.linspace(1e-3, np.pi - 1e-3, n_v)
    P = np.zeros((n_u, n_v, 3))
    N = np.zeros((n_u, n_v, 3))
    UV = np.zeros((n_u, n_v, 2))
    for i, a in enumerate(th):
        N[i, :, 0] = np.sin(phi) * np.cos(a)
        N[i, :, 1] = np.cos(phi)
        N[i, :, 2] = np.sin(phi) * np.sin(a)
        P[i] = N[i] * r + np.asarray(center)
        UV[i, :, 0] = a / (2 * np.pi)
        UV[i, :, 1] = phi / np.pi
    return grid_mesh(P, N, UV, close_u=True)


def box(lo, hi, uv_scale=1.0):
    lo = np.asarray(lo, np.float64); hi = np.asarray(hi, np.float64)
    P = np.zeros((0, 2, 3)); N = np.zeros((0, 2, 3)); UV = np.zeros((0, 2, 2))
    groups = []
    for axis in range(3):
        for s in (0, 1):
            a, b = (axis + 1) % 3, (axis + 2) % 3
            if s == 0:
                a, b = b, a
            quadP = np.zeros((2, 2, 3))
            for ia in range(2):
                for ib in range(2):
                    p = lo.copy()
                    p[axis] = hi[axis] if s else lo[axis]
                    p[a] = hi[a] if ia else lo[a]
                    p[b] = hi[b] if ib else lo[b]
                    quadP[ia, ib] = p
            n = np.zeros(3); n[axis] = 1.0 if s else -1.0
            quadN = np.tile(n, (2, 2, 1))
            quadUV = np.zeros((2, 2, 2))
            for ia in range(2):
                for ib in range(2):
                    quadUV[ia, ib] = (ia * uv_scale, ib * uv_scale)
            groups.append((quadP, quadN, quadUV))
    return groups


# ----------------------------------------------------------------- textures
def write_textures():
    from hippt.utils.png import write_png
    rng = np.random.default_rng(5)
    # wood: ring pattern + noise
    h = w = 256
    y, x = np.mgrid[0:h, 0:w] / h
    rings = np.sin((x * 3 + 0.15 * np.sin(y * 21)) * 40) * 0.5 + 0.5
    grain = rng.random((h, w)) * 0.12
    base = np.stack([0.45 + 0.25 * rings, 0.28 + 0.16 * rings, 0.14 + 0.07 * rings], -1)
    wood = np.clip(base + grain[..., None], 0, 1)
    write_png(os.path.join(HERO, "wood.png"), (wood * 255).astype(np.uint8))
    # ceramic checker glaze
    cells = ((x * 12).astype(int) + (y * 12).astype(int)) % 2
    glaze = np.where(cells[..., None] > 0,
                     np.array([0.82, 0.76, 0.66]), np.array([0.28, 0.42, 0.5]))
    write_png(os.path.join(HERO, "glaze.png"), (glaze * 255).astype(np.uint8))


MTL = """# hero.mtl — four-material still life
newmtl ceramic
Kd 0.85 0.82 0.78
Ks 0.04 0.04 0.04
Ns 40
illum 2
map_Kd glaze.png

newmtl metal
Kd 0.05 0.05 0.05
Ks 0.95 0.64 0.54
Ns 600
illum 3

newmtl glass
Kd 0.0 0.0 0.0
Ks 1.0 1.0 1.0
d 0.2
Ni 1.5
illum 7

newmtl wood
Kd 0.7 0.6 0.5
Ks 0.0 0.0 0.0
Ns 5
illum 2
map_Kd wood.png
"""


def main():
    os.makedirs(HERO, exist_ok=True)
    write_textures()
    with open(os.path.join(HERO, "hero.mtl"), "w") as f:
        f.write(MTL)
    with open(os.path.join(HERO, "hero.obj"), "w") as f:
        f.write("# hero still life (procedural; scripts/make_hero.py)\n")
        f.write("mtllib hero.mtl\n")
        voff = toff = noff = 0
        # vase (ceramic): lathe profile
        prof = []
        for t in np.linspace(0, 1, 40):
            r = 0.16 + 0.12 * np.sin(t * np.pi) - 0.10 * t ** 3 + 0.02 * np.sin(t * 9)
            prof.append((max(r, 0.035), 0.62 * t))
        f.write("usemtl ceramic\n")
        P, N, UV, faces = lathe(prof, n_seg=72, center=(-0.42, 0.12, 0.1),
                                uv_scale=(3.0, 1.0))
        voff, toff, noff = emit_group(f, P, N, UV, faces, voff, toff, noff)
        # metal torus knot
        f.write("usemtl metal\n")
        P, N, UV, faces = torus_knot(center=(0.38, 0.315, 0.0))
        voff, toff, noff = emit_group(f, P, N, UV, faces, voff, toff, noff)
        # glass orb
        f.write("usemtl glass\n")
        P, N, UV, faces = uv_sphere(r=0.17, center=(0.0, 0.345, -0.32))
        voff, toff, noff = emit_group(f, P, N, UV, faces, voff, toff, noff)
        # wood pedestal
        f.write("usemtl wood\n")
        for quadP, quadN, quadUV in box((-0.85, 0.0, -0.65), (0.85, 0.12, 0.55),
                                        uv_scale=2.0):
            _, _, _, faces = grid_mesh(quadP, quadN, quadUV)
            voff, toff, noff = emit_group(f, quadP, quadN, quadUV, faces,
                                          voff, toff, noff)
    print("hero asset written to", HERO)


if __name__ == "__main__":
    main()
