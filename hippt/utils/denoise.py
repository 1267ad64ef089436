"""Edge-aware a-trous (SVGF-lite) denoiser over the renderer's primary-hit
AOV guides (normal/depth/albedo).  Pure torch ops, so it runs on the GPU
tensors the renderer already owns — no training, no external weights.
Extension beyond the reference (which ships no denoiser)."""
from __future__ import annotations


def atrous_denoise(color, normal, depth, albedo, iterations: int = 3,
                   sigma_n: float = 16.0, sigma_z: float = 1.0,
                   sigma_l: float = 4.0):
    """color (h,w,3), normal (h,w,3), depth (h,w), albedo (h,w,3) — torch
    tensors (any device) or numpy arrays.  Returns same-type (h,w,3)."""
    import torch

    was_np = not torch.is_tensor(color)
    if was_np:
        import numpy as np
        color = torch.from_numpy(np.ascontiguousarray(color))
        normal = torch.from_numpy(np.ascontiguousarray(normal))
        depth = torch.from_numpy(np.ascontiguousarray(depth))
        albedo = torch.from_numpy(np.ascontiguousarray(albedo))

    eps = 1e-4
    # normalize guide normals; pixels whose mean normal is degenerate (mixed
    # front/back hits at silhouettes) opt out of normal edge-stopping rather
    # than zeroing every weight
    nn = normal
    nlen = nn.pow(2).sum(dim=-1, keepdim=True).sqrt()
    n_ok = (nlen > 0.3).float()
    normal = nn / nlen.clamp(min=1e-6)
    # demodulate albedo so texture detail is not blurred
    alb = albedo.clamp(min=eps)
    x = (color / alb).permute(2, 0, 1)          # (3,h,w)
    n = normal.permute(2, 0, 1)
    nok = n_ok.permute(2, 0, 1)
    z = depth[None]
    # 5-tap B3 spline offsets/weights per axis -> 25 taps separably applied
    offs = [-2, -1, 0, 1, 2]
    wk = [1 / 16, 1 / 4, 3 / 8, 1 / 4, 1 / 16]

    def shift(t, dy, dx):
        return torch.roll(t, shifts=(dy, dx), dims=(-2, -1))

    for it in range(iterations):
        step = 1 << it
        acc = torch.zeros_like(x)
        wsum = torch.zeros_like(z)
        lum = x.mean(dim=0, keepdim=True)
        for iy, oy in enumerate(offs):
            for ix, ox in enumerate(offs):
                dy, dx = oy * step, ox * step
                k = wk[iy] * wk[ix]
                nq = shift(n, dy, dx)
                zq = shift(z, dy, dx)
                lq = shift(lum, dy, dx)
                nokq = shift(nok, dy, dx)
                w_n = (n * nq).sum(dim=0, keepdim=True).clamp(min=0.0) ** sigma_n
                w_n = w_n * nok * nokq + (1.0 - nok * nokq)  # degenerate -> 1
                w_z = torch.exp(-(z - zq).abs() / (sigma_z * step + eps))
                w_l = torch.exp(-(lum - lq).abs() / (sigma_l + eps))
                w = k * w_n * w_z * w_l
                acc = acc + shift(x, dy, dx) * w
                wsum = wsum + w
        x = acc / wsum.clamp(min=1e-12)
    out = (x.permute(1, 2, 0) * alb).contiguous()
    if was_np:
        return out.numpy()
    return out
