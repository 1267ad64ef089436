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
    def render(self, spp: int = 1):
        """Accumulate spp samples; returns deep-copied (H,W,4) torch tensor
        with mean radiance RGB + accumulated spp in alpha."""
        self._check()
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

    def _check(self):
        if self._released:
            raise RuntimeError("renderer already released")
