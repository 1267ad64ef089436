"""Renderer frontends: megakernel PT, wavefront PT, volumetric PT, light
tracer (optionally bidirectional), depth and BVH-cost visualizers.

Capability parity: reference src/renderer/tracer_base.cuh (render /
render_online / render_raw / variance buffer / update_camera / param_setter)
+ the concrete renderer classes of src/renderer/*.cuh.  GPU accumulation
buffers are PyTorch-ROCm tensors so `raw()` is a zero-copy view for DDP
all-reduce; the CPU path uses numpy and the same native integrator.
"""
from __future__ import annotations

import time
from typing import Optional

import numpy as np

from .. import C
from ..scene.scene import Scene
from ..utils.png import tonemap, write_png

# On the CPU reference path, "wfpt" and "pt-dyn" run the same single-source
# integrator as "pt" (the wavefront pipeline and persistent scheduler are
# GPU execution strategies, not different estimators) — GPU wfpt numerics
# tests compare against exactly this estimator.
RENDERER_IDS = {
    "pt": C.R_MEGAKERNEL_PT,
    "pt-dyn": C.R_MEGAKERNEL_PT_DYN,  # persistent-tile scheduler (GPU)
    "wfpt": C.R_WAVEFRONT_PT,
    "vpt": C.R_VOLUME_PT,
    "lt": C.R_LIGHT_TRACE,
    "bdpt": C.R_LIGHT_TRACE,         # light tracing w/ interleaved PT pass
    "depth": C.R_DEPTH,
    "bvh-cost": C.R_BVH_COST,
}


class FrameTimer:
    """Sliding-window frame timer (reference python_render.cuh:33-56)."""

    def __init__(self, window=32):
        self.window = window
        self.times = []

    def add(self, ms: float):
        self.times.append(ms)
        if len(self.times) > self.window:
            self.times.pop(0)

    def avg(self) -> float:
        return float(np.mean(self.times)) if self.times else 0.0


class Renderer:
    def __init__(self, scene: Scene, device: Optional[int] = None, seed_offset: int = 0):
        self.scene = scene
        cfg = scene.desc.config
        self.kind = cfg.renderer
        self.rid = RENDERER_IDS[self.kind]
        self.seed_offset = int(seed_offset)
        self.spec_constraint = cfg.spec_constraint
        self.caustic_scaling = cfg.caustic_scaling
        self.bidirectional = cfg.bidirectional or self.kind == "bdpt"
        if cfg.use_tof and self.kind not in ("pt", "pt-dyn", "vpt"):
            # wavefront / light tracing do not track path time; gating
            # min_time > 0 there would silently produce black frames
            raise ValueError(
                f"ToF (transient) rendering supports pt/pt-dyn/vpt, not "
                f"{self.kind!r} (reference scope: vpt, megakernel_vpt.cu)")
        self.accum_cnt = 0
        self.timer = FrameTimer()
        self.device = device
        w, h = scene.width, scene.height
        if device is not None:
            import torch
            if not torch.cuda.is_available():
                raise RuntimeError(
                    "GPU renderer requested but torch.cuda.is_available() is False; "
                    "the hippt HIP path refuses to fall back silently")
            torch.cuda.set_device(device)
            C.dev_set_device(device)
            self.torch = torch
            self.accum = torch.zeros((h, w, 4), dtype=torch.float32, device=f"cuda:{device}")
            self.var = torch.zeros((h, w, 2), dtype=torch.float32, device=f"cuda:{device}")
            scene.upload(device)
        else:
            self.torch = None
            self.accum = np.zeros((h, w, 4), np.float32)
            self.var = np.zeros((h, w, 2), np.float32)

    # ------------------------------------------------------------ rendering
    def _seed(self) -> int:
        return (self.seed_offset * 4201) & 0xFFFFFFFF

    def render_adaptive(self, spp: int, batch: int = 0):
        """Variance-guided adaptive sampling (extension beyond the
        reference): the Welford variance buffer is routed back into a
        per-pixel sample budget, n_i ∝ σ_i (equal-error allocation).  The
        first batch samples uniformly; later batches concentrate on noisy
        pixels.  Exact estimator: the accumulator's alpha channel carries
        per-pixel counts.  Megakernel renderers only."""
        if self.rid not in (C.R_MEGAKERNEL_PT, C.R_VOLUME_PT):
            raise ValueError("adaptive sampling needs a megakernel renderer")
        batch = batch if batch > 0 else max(4, spp // 4)
        first = min(batch, spp)
        self.render(first)
        done = first
        while done < spp:
            b = min(batch, spp - done)
            m = self._spp_budget(b)
            self.render(b, spp_map=m)
            done += b
        return self

    def _spp_budget(self, batch: int):
        """Per-pixel uint8 budget for `batch` average spp, n_i ∝ σ_i."""
        if self.device is not None:
            t = self.torch
            n = self.accum[:, :, 3].clamp(min=1.0)
            s, s2 = self.var[:, :, 0], self.var[:, :, 1]
            var_mean = ((s2 / n - (s / n) ** 2).clamp(min=0.0) / n)
            w = var_mean.sqrt()
            # 4x4 box smooth for robustness at low counts
            w = t.nn.functional.avg_pool2d(w[None, None], 5, 1, 2)[0, 0]
            w = w / w.mean().clamp(min=1e-12)
            # bound the redistribution: heavy-tailed (firefly) pixels would
            # otherwise capture the whole budget and starve the rest
            w = w.clamp(0.25, 4.0)
            w = w / w.mean().clamp(min=1e-12)
            m = (w * batch).round().clamp(0, 255).to(t.uint8)
            return m.contiguous()
        import numpy as _np
        n = _np.maximum(self.accum[:, :, 3], 1.0)
        s, s2 = self.var[:, :, 0], self.var[:, :, 1]
        var_mean = _np.maximum(s2 / n - (s / n) ** 2, 0.0) / n
        w = _np.sqrt(var_mean)
        k = _np.ones((5, 5), _np.float32) / 25.0
        from numpy.lib.stride_tricks import sliding_window_view
        pad = _np.pad(w, 2, mode="edge")
        w = (sliding_window_view(pad, (5, 5))[:w.shape[0], :w.shape[1]] * k).sum(axis=(2, 3))
        w = w / max(float(w.mean()), 1e-12)
        w = _np.clip(w, 0.25, 4.0)          # bounded redistribution (fireflies)
        w = w / max(float(w.mean()), 1e-12)
        return _np.clip(_np.round(w * batch), 0, 255).astype(_np.uint8)

    def enable_aov(self):
        """Allocate the primary-hit AOV buffer (normal/depth/albedo sums);
        subsequent megakernel renders fill it (denoiser guides)."""
        h, w = self.scene.desc.camera.height, self.scene.desc.camera.width
        if self.device is not None:
            self.aux = self.torch.zeros((h, w, 8), dtype=self.torch.float32,
                                        device=f"cuda:{self.device}")
        else:
            self.aux = np.zeros((h, w, 8), np.float32)
        return self

    def aov(self):
        """dict(normal (h,w,3), depth (h,w), albedo (h,w,3)) — means."""
        a = self.aux
        cnt = a[:, :, 7:8]
        c = cnt.clip(1e-9, None) if self.device is None else cnt.clamp(min=1e-9)
        return {"normal": a[:, :, 0:3] / c, "depth": a[:, :, 3] / c[:, :, 0],
                "albedo": a[:, :, 4:7] / c}

    def denoise(self, iterations: int = 2, **kw):
        """SVGF-lite a-trous denoise of the current accumulation using the
        AOV guides (enable_aov() first).  Returns (h,w,3)."""
        if getattr(self, "aux", None) is None:
            raise RuntimeError("call enable_aov() before denoise()")
        from ..utils.denoise import atrous_denoise
        img = self.raw()[:, :, :3]
        g = self.aov()
        return atrous_denoise(img, g["normal"], g["depth"], g["albedo"],
                              iterations=iterations, **kw)

    def render(self, spp: int = 1, y0: int = 0, y1: int = 0, spp_map=None):
        """Accumulate spp more samples (reference render_raw semantics).
        y0/y1 restrict rendering to the row band [y0, y1) — tile-split DP
        (megakernel renderers only; 0,0 = full frame)."""
        if self.rid not in (C.R_MEGAKERNEL_PT, C.R_MEGAKERNEL_PT_DYN, C.R_VOLUME_PT,
                            C.R_DEPTH, C.R_BVH_COST):
            # frame-wide pipelines (wavefront, light tracing) would silently
            # ignore per-pixel budgets / row bands — refuse instead
            if spp_map is not None:
                raise ValueError(f"spp_map (adaptive) unsupported for {self.kind!r}")
            if (y0, y1) != (0, 0):
                raise ValueError(f"row-band rendering unsupported for {self.kind!r}")
        t0 = time.perf_counter()
        if self.device is not None:
            stream = self.torch.cuda.current_stream().cuda_stream
            if self.bidirectional:
                # interleave a PT pass and an LT pass (reference light_tracer.cu:43-59)
                self.scene.native.render_device(
                    self.accum.data_ptr(), self.var.data_ptr(), self.accum_cnt, spp,
                    self._seed(), C.R_MEGAKERNEL_PT, 0, 1.0, stream)
                self.scene.native.render_device(
                    self.accum.data_ptr(), 0, self.accum_cnt, spp,
                    self._seed() + 0x9E37, C.R_LIGHT_TRACE,
                    self.spec_constraint, self.caustic_scaling, stream)
            else:
                aux = getattr(self, "aux", None)
                self.scene.native.render_device(
                    self.accum.data_ptr(), self.var.data_ptr(), self.accum_cnt, spp,
                    self._seed(), self.rid, self.spec_constraint, self.caustic_scaling, stream,
                    y0, y1, 0 if spp_map is None else spp_map.data_ptr(),
                    0 if aux is None else aux.data_ptr())
            self.torch.cuda.synchronize(self.device)
        else:
            var = self.var.reshape(-1)
            if self.bidirectional:
                self.scene.native.render_host(self.accum.reshape(-1), self.var.reshape(-1),
                                              self.accum_cnt, spp, self._seed(),
                                              C.R_MEGAKERNEL_PT, 0, 1.0, 0)
                self.scene.native.render_host(self.accum.reshape(-1), None,
                                              self.accum_cnt, spp, self._seed() + 0x9E37,
                                              C.R_LIGHT_TRACE, self.spec_constraint,
                                              self.caustic_scaling, 0)
            else:
                aux = getattr(self, "aux", None)
                self.scene.native.render_host(self.accum.reshape(-1), self.var.reshape(-1),
                                              self.accum_cnt, spp, self._seed(),
                                              self.rid, self.spec_constraint,
                                              self.caustic_scaling, 0, y0, y1,
                                              None if spp_map is None else spp_map.reshape(-1),
                                              aux.reshape(-1) if aux is not None else None)
        if spp_map is None:
            self.accum_cnt += spp
        else:
            # heterogeneous budgets: advance the sampler-stream base past the
            # largest per-pixel count so no pixel reuses a sample index
            m = int(spp_map.max())
            self.accum_cnt += max(spp, m)
        self.timer.add((time.perf_counter() - t0) * 1000.0)
        return self

    def raw(self):
        """Mean radiance image (h,w,4): RGB + spp count in alpha."""
        if self.device is not None:
            cnt = self.accum[:, :, 3:4].clamp(min=1e-9)
            out = self.accum.clone()
            out[:, :, :3] /= cnt
            return out
        cnt = np.maximum(self.accum[:, :, 3:4], 1e-9)
        out = self.accum.copy()
        out[:, :, :3] /= cnt
        return out

    def variance(self):
        """Per-pixel variance of the mean luminance estimate (h,w,1).
        Uses the PER-PIXEL sample count (accumulator alpha) so it stays
        correct under adaptive sampling and band rendering."""
        if self.device is not None:
            n = self.accum[:, :, 3].clamp(min=1.0)
            s, s2 = self.var[:, :, 0], self.var[:, :, 1]
            v = (s2 - s * s / n) / (n - 1).clamp(min=1.0) / n
            return v.clamp(min=0).unsqueeze(-1)
        n = np.maximum(self.accum[:, :, 3], 1.0)
        s, s2 = self.var[:, :, 0], self.var[:, :, 1]
        v = (s2 - s * s / n) / np.maximum(n - 1, 1.0) / n
        return np.clip(v, 0, None)[:, :, None]

    # ----------------------------------------------------------- utilities
    def reset(self):
        if self.device is not None:
            self.accum.zero_()
            self.var.zero_()
        else:
            self.accum[:] = 0
            self.var[:] = 0
        self.accum_cnt = 0

    def update_camera(self, **kw):
        # no device re-upload needed: render_device refreshes the camera
        # from the holder on every call (hot-reload path)
        self.scene.update_camera(**kw)
        self.reset()

    def counter(self) -> int:
        return self.accum_cnt

    def avg_frame_time(self) -> float:
        return self.timer.avg()

    def save(self, path: str, gamma: float = 2.1, exposure: float = 1.0):
        acc = self.accum.cpu().numpy() if self.device is not None else self.accum
        write_png(path, tonemap(acc, gamma, exposure))

    def image(self, gamma: float = 2.1, exposure: float = 1.0) -> np.ndarray:
        acc = self.accum.cpu().numpy() if self.device is not None else self.accum
        return tonemap(acc, gamma, exposure)
