"""Wavefront OBJ loader (numpy-based).

Capability parity: reference uses tinyobjloader (scene.cu:548-660) including
shading-normal passthrough, UVs, and per-face materials; this loader handles
v/vn/vt/f with arbitrary polygon fan triangulation, negative indices,
`usemtl` face groups, and `.mtl` material libraries (load_obj_multi /
load_mtl) for multi-material hero assets (scenes/hero.xml).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import numpy as np


def load_mtl(path: str) -> Dict[str, dict]:
    """Parse a .mtl library -> {name: {kd, ks, ns, d, ni, map_kd, illum}}.
    Colors are float triples; map_kd is a path relative to the mtl file."""
    mats: Dict[str, dict] = {}
    cur = None
    base = os.path.dirname(path)
    with open(path, "r", errors="ignore") as f:
        for line in f:
            p = line.split()
            if not p:
                continue
            k = p[0].lower()
            if k == "newmtl":
                cur = {"kd": (0.8, 0.8, 0.8), "ks": (0.0, 0.0, 0.0), "ns": 0.0,
                       "d": 1.0, "ni": 1.5, "map_kd": None, "illum": 2}
                mats[p[1]] = cur
            elif cur is None:
                continue
            elif k == "kd":
                cur["kd"] = tuple(float(v) for v in p[1:4])
            elif k == "ks":
                cur["ks"] = tuple(float(v) for v in p[1:4])
            elif k == "ns":
                cur["ns"] = float(p[1])
            elif k == "d":
                cur["d"] = float(p[1])
            elif k == "tr":
                cur["d"] = 1.0 - float(p[1])
            elif k == "ni":
                cur["ni"] = float(p[1])
            elif k == "illum":
                cur["illum"] = int(p[1])
            elif k == "map_kd":
                cur["map_kd"] = os.path.normpath(os.path.join(base, p[-1]))
    return mats


def _parse(path: str):
    """Shared OBJ scan -> (v, vn, vt, faces, face_mtl, mtllib)."""
    vs, vns, vts = [], [], []
    faces = []  # list of [(vi, ti, ni), ...]
    face_mtl: List[str] = []
    mtllib = None
    cur_mtl = ""
    with open(path, "r", errors="ignore") as f:
        for line in f:
            if line.startswith("v "):
                p = line.split()
                vs.append((float(p[1]), float(p[2]), float(p[3])))
            elif line.startswith("vn "):
                p = line.split()
                vns.append((float(p[1]), float(p[2]), float(p[3])))
            elif line.startswith("vt "):
                p = line.split()
                vts.append((float(p[1]), float(p[2])))
            elif line.startswith("usemtl"):
                cur_mtl = line.split(maxsplit=1)[1].strip() if " " in line else ""
            elif line.startswith("mtllib"):
                mtllib = os.path.normpath(os.path.join(
                    os.path.dirname(path), line.split(maxsplit=1)[1].strip()))
            elif line.startswith("f "):
                p = line.split()[1:]
                idx = []
                for tok in p:
                    parts = tok.split("/")
                    vi = int(parts[0])
                    ti = int(parts[1]) if len(parts) > 1 and parts[1] else 0
                    ni = int(parts[2]) if len(parts) > 2 and parts[2] else 0
                    idx.append((vi, ti, ni))
                for k in range(1, len(idx) - 1):  # fan triangulation
                    faces.append([idx[0], idx[k], idx[k + 1]])
                    face_mtl.append(cur_mtl)
    return vs, vns, vts, faces, face_mtl, mtllib


def _assemble(vs, vns, vts, faces):
    v = np.asarray(vs, np.float32)
    vn = np.asarray(vns, np.float32) if vns else None
    vt = np.asarray(vts, np.float32) if vts else None

    def fix(i, n):
        return i - 1 if i > 0 else n + i

    n_f = len(faces)
    tris = np.zeros((n_f, 3, 3), np.float32)
    normals = np.zeros((n_f, 3, 3), np.float32) if vn is not None else None
    uvs = np.zeros((n_f, 3, 2), np.float32) if vt is not None else None
    has_n = has_t = False
    for fi, face in enumerate(faces):
        for ci, (vi, ti, ni) in enumerate(face):
            tris[fi, ci] = v[fix(vi, len(v))]
            if normals is not None and ni != 0:
                normals[fi, ci] = vn[fix(ni, len(vn))]
                has_n = True
            if uvs is not None and ti != 0:
                uvs[fi, ci] = vt[fix(ti, len(vt))]
                has_t = True
    return tris, (normals if has_n else None), (uvs if has_t else None)


def load_obj_multi(path: str):
    """Load an OBJ split by `usemtl` groups.  Returns (groups, materials)
    where groups = [(mtl_name, tris, normals|None, uvs|None), ...] in first-
    use order and materials = the parsed mtllib dict (may be empty)."""
    vs, vns, vts, faces, face_mtl, mtllib = _parse(path)
    materials = load_mtl(mtllib) if mtllib and os.path.exists(mtllib) else {}
    if not faces:
        return [], materials
    order: List[str] = []
    for m in face_mtl:
        if m not in order:
            order.append(m)
    groups = []
    for name in order:
        sub = [f for f, m in zip(faces, face_mtl) if m == name]
        groups.append((name, *_assemble(vs, vns, vts, sub)))
    return groups, materials


def load_obj(path: str) -> Tuple[np.ndarray, Optional[np.ndarray], Optional[np.ndarray]]:
    """Returns (tris (n,3,3), normals (n,3,3)|None, uvs (n,3,2)|None)."""
    vs, vns, vts, faces, _face_mtl, _mtllib = _parse(path)
    if not faces:
        return np.zeros((0, 3, 3), np.float32), None, None
    return _assemble(vs, vns, vts, faces)


def save_obj(path: str, tris: np.ndarray) -> None:
    """Write a triangle soup as OBJ (test fixture generation)."""
    tris = np.asarray(tris, np.float32).reshape(-1, 3, 3)
    with open(path, "w") as f:
        for t in tris:
            for p in t:
                f.write(f"v {p[0]} {p[1]} {p[2]}\n")
        for i in range(len(tris)):
            b = 3 * i
            f.write(f"f {b + 1} {b + 2} {b + 3}\n")
