"""False-color maps for the depth / BVH-cost debug renderers.

Capability parity: reference src/impl/color_map.cu (PLASMA/JET/VIRIDIS
256-entry tables as 1D textures) + depth.cu false_color_mapping (min/max
normalization, optional log transform).  Implemented as compact polynomial
fits instead of shipped tables.
"""
from __future__ import annotations

import numpy as np


def _viridis(t):
    # polynomial fit of matplotlib viridis (public domain endpoints)
    c0 = np.array([0.2777, 0.0054, 0.3340])
    c1 = np.array([0.1050, 1.4046, 1.3845])
    c2 = np.array([-0.3308, 0.2148, 0.0950])
    c3 = np.array([-4.6342, -5.7991, -19.3324])
    c4 = np.array([6.2282, 14.1799, 56.6905])
    c5 = np.array([4.7763, -13.7451, -65.3530])
    c6 = np.array([-5.4354, 4.6458, 26.3124])
    t = t[..., None]
    return c0 + t * (c1 + t * (c2 + t * (c3 + t * (c4 + t * (c5 + t * c6)))))


def _plasma(t):
    c0 = np.array([0.0504, 0.0298, 0.5280])
    c1 = np.array([2.1766, 0.2383, 0.7539])
    c2 = np.array([-2.6894, -7.4558, 3.1107])
    c3 = np.array([6.1303, 42.3461, -28.5188])
    c4 = np.array([-11.1074, -82.6663, 60.1398])
    c5 = np.array([10.0230, 71.4136, -54.0722])
    c6 = np.array([-3.6587, -22.9315, 18.1919])
    t = t[..., None]
    return c0 + t * (c1 + t * (c2 + t * (c3 + t * (c4 + t * (c5 + t * c6)))))


def _jet(t):
    r = np.clip(1.5 - np.abs(4 * t - 3), 0, 1)
    g = np.clip(1.5 - np.abs(4 * t - 2), 0, 1)
    b = np.clip(1.5 - np.abs(4 * t - 1), 0, 1)
    return np.stack([r, g, b], axis=-1)


COLOR_MAPS = {"viridis": _viridis, "plasma": _plasma, "jet": _jet}


def false_color(values: np.ndarray, cmap: str = "plasma", log_scale: bool = False,
                vmin=None, vmax=None) -> np.ndarray:
    """values (h,w) -> (h,w,3) float in [0,1]; zeros (misses) map to black."""
    v = np.asarray(values, np.float64).copy()
    mask = v > 0
    if log_scale:
        v[mask] = np.log2(v[mask] + 1.0)
    lo = v[mask].min() if vmin is None and mask.any() else (vmin or 0.0)
    hi = v[mask].max() if vmax is None and mask.any() else (vmax or 1.0)
    t = np.zeros_like(v)
    if hi > lo:
        t[mask] = (v[mask] - lo) / (hi - lo)
    rgb = np.clip(COLOR_MAPS[cmap](t), 0, 1)
    rgb[~mask] = 0.0
    return rgb.astype(np.float32)
