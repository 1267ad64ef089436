"""Wavefront OBJ loader (numpy-based).

Capability parity: reference uses tinyobjloader (scene.cu:548-660) including
shading-normal passthrough and UVs; this loader handles v/vn/vt/f with
arbitrary polygon fan triangulation and negative indices.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np


def load_obj(path: str) -> Tuple[np.ndarray, Optional[np.ndarray], Optional[np.ndarray]]:
    """Returns (tris (n,3,3), normals (n,3,3)|None, uvs (n,3,2)|None)."""
    vs, vns, vts = [], [], []
    faces = []  # list of [(vi, ti, ni), ...]
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
    if not faces:
        return np.zeros((0, 3, 3), np.float32), None, None
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
