"""pyrender-compatible PythonRenderer.

Capability parity: reference src/pyrender/python_render.cu + python_bind.cu —
`PythonRenderer(xml, device_id, seed_offset)` with `render()` returning a
deep-copied (H, W, 4) float32 torch tensor on the selected device,
`variance()` (H, W, 1), `counter()`, `avg_frame_time()`, `info()`,
`release()`.  Additionally accepts a SceneDesc directly (procedural scenes)
and `device_id=-1` for the CPU reference path (BASELINE config #1).
"""
from __future__ import annotations

import os
from typing import Optional, Union

from .scene.scene import Scene, SceneDesc


class PythonRenderer:
    def __init__(self, scene: Union[str, SceneDesc], device_id: int = 0,
                 seed_offset: int = 0, renderer: Optional[str] = None):
        if isinstance(scene, str):
            from .scene.xml_parser import parse_xml
            desc = parse_xml(scene)
        else:
            desc = scene
        if renderer is not None:
            desc.config.renderer = renderer
        self.desc = desc
        self.device_id = device_id
        use_gpu = device_id is not None and device_id >= 0
        if use_gpu:
            import torch
            if not torch.cuda.is_available():
                raise RuntimeError(
                    "PythonRenderer: device_id >= 0 requires a visible GPU "
                    "(pass device_id=-1 for the CPU reference path)")
        from .render.renderer import Renderer
        self.scene = Scene(desc)
        self.renderer = Renderer(self.scene, device=device_id if use_gpu else None,
                                 seed_offset=seed_offset)
        self._released = False

    # -- reference API ------------------------------------------------------
    def render(self, spp: int = 1, adaptive: bool = False):
        """Accumulate spp samples; returns deep-copied (H,W,4) torch tensor
        with mean radiance RGB + accumulated spp in alpha.  adaptive=True
        routes the variance buffer into per-pixel sample budgets."""
        self._check()
        if adaptive:
            self.renderer.render_adaptive(spp)
        else:
            self.renderer.render(spp)
        raw = self.renderer.raw()
        if self.renderer.device is not None:
            return raw  # raw() already deep-copies (clone + divide)
        import torch
        return torch.from_numpy(raw.copy())

    def variance(self):
        self._check()
        v = self.renderer.variance()
        if self.renderer.device is not None:
            return v
        import torch
        return torch.from_numpy(v.copy())

    def counter(self) -> int:
        return self.renderer.counter()

    def avg_frame_time(self) -> float:
        return self.renderer.avg_frame_time()

    def info(self) -> dict:
        return self.scene.info()

    def release(self):
        if not self._released:
            self.scene.native.release()
            self._released = True

    # -- extras -------------------------------------------------------------
    def reset(self):
        self.renderer.reset()

    def update_camera(self, **kw):
        self.renderer.update_camera(**kw)

    def save(self, path: str, gamma: float = 2.1):
        self.renderer.save(path, gamma)

    def enable_aov(self):
        """Allocate primary-hit AOV buffers (denoiser guides)."""
        self.renderer.enable_aov()
        return self

    def aov(self):
        """dict(normal, depth, albedo) means over accumulated samples."""
        return self.renderer.aov()

    def denoise(self, iterations: int = 2, **kw):
        """SVGF-lite a-trous denoise of the accumulation (enable_aov first);
        returns an (H, W, 3) tensor/array on the render device."""
        return self.renderer.denoise(iterations=iterations, **kw)

    def save_state(self, path: str, **extra):
        """Checkpoint the warm accumulation state (radiance sums, variance
        sums, sample counter) so a long accumulation can resume after a
        restart.  The reference keeps this state implicit and unpersisted
        (tracer_base.cuh:135-158 accum buffer + accum_cnt); here it is a
        first-class .npz snapshot.  Extra scalar/array keys (e.g. the driver
        loop's step index) ride along and come back from load_state()."""
        self._check()
        import numpy as np
        r = self.renderer
        accum = r.accum.cpu().numpy() if r.device is not None else r.accum
        var = r.var.cpu().numpy() if r.device is not None else r.var
        np.savez_compressed(path, accum=accum, var=var,
                            accum_cnt=np.int64(r.accum_cnt), **extra)

    def load_state(self, path: str):
        """Resume from a save_state() snapshot (shape-checked).  Returns the
        dict of extra keys stored by save_state (empty if none)."""
        self._check()
        import numpy as np
        z = np.load(path)
        r = self.renderer
        if tuple(z["accum"].shape) != tuple(r.accum.shape):
            raise ValueError(f"checkpoint shape {z['accum'].shape} != "
                             f"framebuffer {tuple(r.accum.shape)}")
        if r.device is not None:
            import torch
            r.accum.copy_(torch.from_numpy(z["accum"]).to(r.accum.device))
            r.var.copy_(torch.from_numpy(z["var"]).to(r.var.device))
        else:
            r.accum[:] = z["accum"]
            r.var[:] = z["var"]
        r.accum_cnt = int(z["accum_cnt"])
        return {k: z[k] for k in z.files if k not in ("accum", "var", "accum_cnt")}

    def _check(self):
        if self._released:
            raise RuntimeError("renderer already released")
