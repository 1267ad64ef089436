"""Interactive viewer with online parameter hot-reload.

Capability parity: reference `cpt` viewer (app/viewer.cu + viewer_impl/
imgui_utils.cu): progressive online rendering, WASD/mouse camera, live
BSDF/emitter/medium parameter editing with accumulation reset on change,
renderer switching, frame capture, FPS readout.

MI355X-native substitution: the reference renders into a GL PBO via CUDA-GL
interop under GLFW/ImGui.  MI355X nodes are headless (no GL/display), so the
idiomatic equivalent is a web viewer: a FastAPI app streams the progressive
framebuffer over a client-paced BINARY WEBSOCKET (/ws/stream: 16-byte
header + tonemapped RGB, drawn into a canvas via ImageData — no
per-request PNG encode on the hot path; /frame.png stays for
capture/fallback) plus JSON control endpoints, and a background thread
keeps accumulating samples (the render loop).  The browser page implements
WASD/mouse camera and parameter panels.  Everything is testable headlessly
through the HTTP API (tests/test_viewer.py).
"""
from __future__ import annotations

import io
import threading
import time
from typing import Optional

import numpy as np

# module-level so FastAPI can resolve the postponed `ws: WebSocket`
# annotation (from __future__ import annotations stringifies it and the
# lookup happens in module globals); guarded: the viewer is optional.
try:
    from fastapi import WebSocket
except Exception:  # pragma: no cover - fastapi always ships in this image
    WebSocket = None


INDEX_HTML = """<!doctype html>
<html><head><title>hippt viewer</title>
<style>
 body { background:#151515; color:#ddd; font-family:monospace; margin:0; display:flex }
 #img { image-rendering:pixelated; }
 #panel { padding:12px; width:320px; font-size:12px }
 input,select,button { background:#222; color:#ddd; border:1px solid #555; margin:2px }
 .row { margin:4px 0 }
</style></head>
<body>
<div><canvas id="cv" tabindex="0"></canvas><img id="img" style="display:none"></div>
<div id="panel">
  <div class="row">fps: <span id="fps">-</span> spp: <span id="spp">-</span></div>
  <div class="row">scene:
    <select id="scene" onchange="fetch('/api/scene',{method:'POST',
      headers:{'Content-Type':'application/json'},
      body:JSON.stringify({scene:this.value})})">
      <option>cornell</option><option>kitchen</option>
      <option>sports-car</option><option>smoke</option>
    </select></div>
  <div class="row">renderer:
    <select id="renderer">
      <option>pt</option><option>pt-dyn</option><option>wfpt</option><option>vpt</option>
      <option>lt</option><option>bdpt</option><option>depth</option><option>bvh-cost</option>
    </select></div>
  <div class="row">bsdf: <select id="bsdf_i"></select>
    type: <select id="bsdf_t">
      <option>lambertian</option><option>specular</option><option>glass</option>
      <option>plastic</option><option>plastic-forward</option><option>ggx</option>
      <option>dispersion</option><option>forward</option></select>
    kd: <input id="kd" size="10" value="0.8,0.8,0.8">
    <button onclick="setBsdf()">apply</button></div>
  <div class="row">emitter scale: <input id="escale" size="6" value="20">
    <button onclick="setEmitter()">apply</button></div>
  <div class="row">max depth: <input id="maxd" size="4" value="8">
    <button onclick="setDepth()">apply</button></div>
  <div class="row"><button onclick="capture()">capture png</button>
    <button onclick="fetch('/api/reset',{method:'POST'})">reset</button>
    <label><input type="checkbox" onchange="fetch('/api/adaptive',{method:'POST',
      headers:{'Content-Type':'application/json'},
      body:JSON.stringify({enabled:this.checked})})"> adaptive spp</label>
    <label><input type="checkbox" onchange="fetch('/api/denoise',{method:'POST',
      headers:{'Content-Type':'application/json'},
      body:JSON.stringify({enabled:this.checked})})"> denoise</label></div>
  <div class="row">WASD move, QE up/down, arrows look</div>
</div>
<script>
const cv = document.getElementById('cv');
const ctx = cv.getContext('2d');
// client-paced binary stream: ask for a frame, draw it, ask again.
function connect(){
  const ws = new WebSocket((location.protocol==='https:'?'wss://':'ws://') +
                           location.host + '/ws/stream');
  ws.binaryType = 'arraybuffer';
  ws.onopen = () => ws.send('next');
  ws.onmessage = ev => {
    const dv = new DataView(ev.data);
    const w = dv.getUint32(0, true), h = dv.getUint32(4, true);
    if (cv.width !== w) { cv.width = w; cv.height = h; }
    const rgb = new Uint8Array(ev.data, 16);
    const id = ctx.createImageData(w, h);
    for (let i = 0, j = 0; j < rgb.length; i += 4, j += 3) {
      id.data[i] = rgb[j]; id.data[i+1] = rgb[j+1];
      id.data[i+2] = rgb[j+2]; id.data[i+3] = 255;
    }
    ctx.putImageData(id, 0, 0);
    ws.send('next');
  };
  ws.onclose = () => setTimeout(connect, 800);
}
connect();
setInterval(async () => {
  const s = await (await fetch('/api/stats')).json();
  document.getElementById('fps').textContent = s.fps.toFixed(1);
  document.getElementById('spp').textContent = s.spp;
}, 1000);
fetch('/api/state').then(r=>r.json()).then(s=>{
  const sel = document.getElementById('bsdf_i');
  for (let i=0;i<s.n_bsdfs;i++){const o=document.createElement('option');o.text=i;sel.add(o);}
  document.getElementById('renderer').value = s.renderer;
});
document.getElementById('renderer').onchange = e =>
  fetch('/api/renderer',{method:'POST',headers:{'Content-Type':'application/json'},
        body:JSON.stringify({kind:e.target.value})});
function setBsdf(){
  fetch('/api/bsdf',{method:'POST',headers:{'Content-Type':'application/json'},
    body:JSON.stringify({index:+document.getElementById('bsdf_i').value,
      type:document.getElementById('bsdf_t').value,
      kd:document.getElementById('kd').value.split(',').map(Number)})});
}
function setEmitter(){
  fetch('/api/emitter',{method:'POST',headers:{'Content-Type':'application/json'},
    body:JSON.stringify({index:0,scale:+document.getElementById('escale').value})});
}
function setDepth(){
  fetch('/api/depths',{method:'POST',headers:{'Content-Type':'application/json'},
    body:JSON.stringify({max_depth:+document.getElementById('maxd').value})});
}
function capture(){ window.open('/capture.png'); }
// mouse-drag orbit (reference cpt mouse camera): drag on the canvas
// sends yaw/pitch steps through the same hot-reload camera API
let drag = null;
cv.addEventListener('mousedown', ev => { drag = [ev.clientX, ev.clientY]; });
window.addEventListener('mouseup', () => { drag = null; });
window.addEventListener('mousemove', ev => {
  if (!drag) return;
  const dx = ev.clientX - drag[0], dy = ev.clientY - drag[1];
  if (Math.abs(dx) < 8 && Math.abs(dy) < 8) return;
  drag = [ev.clientX, ev.clientY];
  const keys = [];
  if (dx > 0) keys.push('yaw+'); else if (dx < 0) keys.push('yaw-');
  if (dy > 0) keys.push('pitch-'); else if (dy < 0) keys.push('pitch+');
  for (const k of keys)
    fetch('/api/camera/move',{method:'POST',
      headers:{'Content-Type':'application/json'},body:JSON.stringify({key:k})});
});
window.addEventListener('keydown', ev => {
  const map = {w:'w',a:'a',s:'s',d:'d',q:'q',e:'e',
               ArrowLeft:'yaw-',ArrowRight:'yaw+',ArrowUp:'pitch+',ArrowDown:'pitch-'};
  if (map[ev.key]) fetch('/api/camera/move',{method:'POST',
    headers:{'Content-Type':'application/json'},body:JSON.stringify({key:map[ev.key]})});
});
</script></body></html>
"""


class ViewerApp:
    """Render loop + HTTP control surface."""

    def __init__(self, desc, device: Optional[int] = None, spp_per_frame: int = 1):
        import hippt
        self.pyr = hippt.PythonRenderer(desc, device_id=-1 if device is None else device)
        self.desc = desc
        self.device = device
        self.spp_per_frame = spp_per_frame
        self.lock = threading.Lock()
        self.running = False
        self.thread = None
        self.move_speed = 0.15
        self.rot_speed = 0.08
        self.adaptive = False   # variance-guided per-pixel spp after warmup
        self.denoise = False    # SVGF-lite a-trous display filter

    # ------------------------------------------------------------ lifecycle
    def start(self):
        self.running = True
        self.thread = threading.Thread(target=self._loop, daemon=True)
        self.thread.start()

    def stop(self):
        self.running = False
        if self.thread:
            self.thread.join(timeout=5)

    def _loop(self):
        while self.running:
            with self.lock:
                r = self.pyr.renderer
                if (self.adaptive and r.accum_cnt >= 8
                        and r.rid in (0, 2)):  # megakernel pt/vpt only
                    m = r._spp_budget(self.spp_per_frame)
                    r.render(self.spp_per_frame, spp_map=m)
                else:
                    r.render(self.spp_per_frame)
            time.sleep(0.0005)

    # ------------------------------------------------------------- frames
    def frame_raw(self, scale: int = 1) -> bytes:
        """Tonemapped RGB frame as a binary packet: 16-byte header
        (u32 width, u32 height, u32 spp, u32 reserved, little-endian) +
        w*h*3 bytes.  This is the websocket streaming path — no PNG/zlib
        work per frame (the round-1 viewer PNG-encoded every poll)."""
        import struct
        from ..utils.png import tonemap
        with self.lock:
            r = self.pyr.renderer
            kind = r.kind
            if self.denoise and getattr(r, "aux", None) is not None:
                den = r.denoise()
                den = den.cpu().numpy() if hasattr(den, "cpu") else np.asarray(den)
                acc = np.concatenate([den, np.ones_like(den[..., :1])], axis=2)
            else:
                acc = (r.accum.cpu().numpy()
                       if r.device is not None else r.accum.copy())
            spp = r.accum_cnt
        img = self._present(acc, kind)
        if scale > 1:
            img = img[::scale, ::scale]
        img = np.ascontiguousarray(img)
        h, w = img.shape[:2]
        return struct.pack("<4I", w, h, int(spp), 0) + img.tobytes()

    def _present(self, acc, kind):
        """accum -> displayable RGB: false color for the debug renderers
        (reference cpt colormaps depth/BVH-cost), tonemap otherwise."""
        from ..utils.png import tonemap
        if kind in ("depth", "bvh-cost"):
            from ..utils.colormap import false_color
            vals = acc[:, :, 0] / np.maximum(acc[:, :, 3], 1e-9)
            img = false_color(vals, cmap="plasma", log_scale=kind == "bvh-cost")
            img = (np.clip(img[..., :3], 0, 1) * 255.0 + 0.5).astype(np.uint8)
            return np.ascontiguousarray(img)
        return tonemap(acc)[..., :3]

    def frame_png(self) -> bytes:
        from ..utils.png import tonemap, write_png
        import tempfile, os
        import numpy as np
        with self.lock:
            r = self.pyr.renderer
            if self.denoise and getattr(r, "aux", None) is not None:
                den = r.denoise()
                den = den.cpu().numpy() if hasattr(den, "cpu") else np.asarray(den)
                acc = np.concatenate([den, np.ones_like(den[..., :1])], axis=2)
            else:
                acc = (r.accum.cpu().numpy()
                       if r.device is not None else r.accum.copy())
            kind = r.kind
        img = self._present(acc, kind)
        buf = io.BytesIO()
        # write_png writes to path; reuse its encoder via temp buffer
        tmp = tempfile.NamedTemporaryFile(suffix=".png", delete=False)
        tmp.close()
        write_png(tmp.name, img)
        data = open(tmp.name, "rb").read()
        os.unlink(tmp.name)
        return data

    # ------------------------------------------------------------- controls
    def reset(self):
        with self.lock:
            self.pyr.renderer.reset()

    def load_scene(self, scene: str):
        """Hot-swap the whole scene at runtime (XML path or procedural name)
        — the reference restarts `cpt` per scene; here the render loop keeps
        running and only the renderer object is rebuilt under the lock."""
        import hippt
        from ..parallel.ddp import load_scene as _load
        import argparse
        ns = argparse.Namespace(scene=scene, width=None, height=None,
                                renderer=None)
        desc = _load(ns)
        new_pyr = hippt.PythonRenderer(
            desc, device_id=-1 if self.device is None else self.device)
        with self.lock:
            old = self.pyr
            self.pyr = new_pyr
            self.desc = desc
            self.adaptive = False
            self.denoise = False
        old.release()

    def set_renderer(self, kind: str):
        with self.lock:
            self.desc.config.renderer = kind
            from ..render.renderer import Renderer
            self.pyr.renderer.kind = kind
            from ..render import renderer as rmod
            self.pyr.renderer.rid = rmod.RENDERER_IDS[kind]
            self.pyr.renderer.bidirectional = kind == "bdpt"
            self.pyr.renderer.reset()

    def set_bsdf(self, index: int, **kw):
        from ..scene.scene import BsdfDesc
        with self.lock:
            b = self.desc.bsdfs[index]
            for k, v in kw.items():
                if v is not None and hasattr(b, k):
                    setattr(b, k, tuple(v) if isinstance(v, list) else v)
            self.pyr.scene.set_bsdf(index, b)
            self.pyr.renderer.reset()

    def set_emitter(self, index: int, **kw):
        with self.lock:
            self.pyr.scene.set_emitter(index, **kw)
            self.pyr.renderer.reset()

    def set_medium(self, index: int, **kw):
        with self.lock:
            self.pyr.scene.set_medium(index, **kw)
            self.pyr.renderer.reset()

    def set_depths(self, **kw):
        with self.lock:
            self.pyr.scene.set_depths(**kw)
            self.pyr.renderer.reset()

    def move_camera(self, key: str):
        import numpy as np
        c = self.desc.camera
        pos = np.asarray(c.pos, np.float64)
        look = np.asarray(c.lookat, np.float64)
        fwd = look - pos
        fwd /= np.linalg.norm(fwd)
        up = np.asarray(c.up, np.float64)
        right = np.cross(up, fwd)
        right /= np.linalg.norm(right)
        d = self.move_speed
        delta = {"w": fwd * d, "s": -fwd * d, "a": -right * d, "d": right * d,
                 "q": up * d, "e": -up * d}.get(key)
        if delta is not None:
            pos += delta
            look += delta
        else:
            # yaw/pitch rotate the forward vector (reference camera rotate)
            import math
            ang = self.rot_speed
            if key == "yaw+":
                R = self._rot(up, ang)
            elif key == "yaw-":
                R = self._rot(up, -ang)
            elif key == "pitch+":
                R = self._rot(right, ang)
            elif key == "pitch-":
                R = self._rot(right, -ang)
            else:
                return
            fwd = R @ fwd
            look = pos + fwd
        with self.lock:
            self.pyr.scene.update_camera(pos=tuple(pos), lookat=tuple(look))
            self.pyr.renderer.reset()

    @staticmethod
    def _rot(axis, ang):
        import math
        axis = axis / np.linalg.norm(axis)
        K = np.array([[0, -axis[2], axis[1]], [axis[2], 0, -axis[0]],
                      [-axis[1], axis[0], 0]])
        return np.eye(3) + math.sin(ang) * K + (1 - math.cos(ang)) * (K @ K)

    def stats(self):
        r = self.pyr.renderer
        ft = r.avg_frame_time()
        return {"fps": 1000.0 / ft if ft > 0 else 0.0, "spp": r.counter(),
                "frame_ms": ft}

    def state(self):
        return {"renderer": self.desc.config.renderer,
                "n_bsdfs": len(self.desc.bsdfs),
                "n_emitters": len(self.desc.emitters),
                "n_media": len(self.desc.media),
                "resolution": [self.desc.camera.width, self.desc.camera.height]}



# Request models at module scope: with `from __future__ import annotations`
# FastAPI resolves string annotations via module globals, so locally-scoped
# models would silently degrade to query parameters.
try:
    from pydantic import BaseModel as _BaseModel

    class BsdfReq(_BaseModel):
        index: int
        type: Optional[str] = None
        kd: Optional[list] = None
        ks: Optional[list] = None
        kg: Optional[list] = None
        ior: Optional[float] = None
        roughness_x: Optional[float] = None
        roughness_y: Optional[float] = None
        metal: Optional[str] = None

    class EmitterReq(_BaseModel):
        index: int
        emission: Optional[list] = None
        scale: Optional[float] = None

    class MediumReq(_BaseModel):
        index: int
        sigma_a: Optional[list] = None
        sigma_s: Optional[list] = None
        scale: Optional[float] = None
        emission_scale: Optional[float] = None

    class DepthReq(_BaseModel):
        radiance_clamp: Optional[float] = None
        max_depth: Optional[int] = None
        max_diffuse: Optional[int] = None
        max_specular: Optional[int] = None
        max_transmit: Optional[int] = None
        max_volume: Optional[int] = None

    class RendererReq(_BaseModel):
        kind: str

    class SceneReq(_BaseModel):
        scene: str

    class AdaptiveReq(_BaseModel):
        enabled: bool = True

    class MoveReq(_BaseModel):
        key: str

    class CamReq(_BaseModel):
        fov: Optional[float] = None
        aperture: Optional[float] = None
        focal_dist: Optional[float] = None
        ortho: Optional[bool] = None
except ImportError:  # viewer optional without fastapi/pydantic
    pass

def build_app(viewer: ViewerApp):
    from fastapi import FastAPI, Response

    app = FastAPI(title="hippt viewer")

    @app.get("/")
    def index():
        return Response(INDEX_HTML, media_type="text/html")

    @app.get("/frame.png")
    def frame():
        return Response(viewer.frame_png(), media_type="image/png")

    @app.websocket("/ws/stream")
    async def ws_stream(ws: WebSocket):
        # client-paced: each received message yields one fresh frame, so a
        # slow client never queues stale frames and a fast one streams at
        # render speed
        await ws.accept()
        try:
            while True:
                msg = await ws.receive_text()
                scale = 2 if msg == "next2" else 1
                await ws.send_bytes(viewer.frame_raw(scale=scale))
        except Exception:
            pass

    @app.get("/capture.png")
    def capture():
        return Response(viewer.frame_png(), media_type="image/png",
                        headers={"Content-Disposition": "attachment; filename=render.png"})

    @app.get("/api/stats")
    def stats():
        return viewer.stats()

    @app.get("/api/state")
    def state():
        return viewer.state()

    @app.post("/api/reset")
    def reset():
        viewer.reset()
        return {"ok": True}

    @app.post("/api/renderer")
    def renderer(req: RendererReq):
        viewer.set_renderer(req.kind)
        return {"ok": True}

    @app.post("/api/scene")
    def scene(req: SceneReq):
        viewer.load_scene(req.scene)
        return {"ok": True, "scene": req.scene}

    @app.post("/api/adaptive")
    def adaptive(req: AdaptiveReq):
        viewer.adaptive = bool(req.enabled)
        return {"ok": True, "adaptive": viewer.adaptive}

    @app.post("/api/denoise")
    def denoise(req: AdaptiveReq):
        if req.enabled and getattr(viewer.pyr.renderer, "aux", None) is None:
            with viewer.lock:
                viewer.pyr.renderer.enable_aov()
        viewer.denoise = bool(req.enabled)
        return {"ok": True, "denoise": viewer.denoise}

    @app.post("/api/bsdf")
    def bsdf(req: BsdfReq):
        kw = req.dict()
        i = kw.pop("index")
        viewer.set_bsdf(i, **{k: v for k, v in kw.items() if v is not None})
        return {"ok": True}

    @app.post("/api/emitter")
    def emitter(req: EmitterReq):
        kw = req.dict()
        i = kw.pop("index")
        viewer.set_emitter(i, **{k: v for k, v in kw.items() if v is not None})
        return {"ok": True}

    @app.post("/api/medium")
    def medium(req: MediumReq):
        kw = req.dict()
        i = kw.pop("index")
        viewer.set_medium(i, **{k: v for k, v in kw.items() if v is not None})
        return {"ok": True}

    @app.post("/api/depths")
    def depths(req: DepthReq):
        viewer.set_depths(**{k: v for k, v in req.dict().items() if v is not None})
        return {"ok": True}

    @app.post("/api/camera/params")
    def camera_params(req: CamReq):
        with viewer.lock:
            cam = viewer.pyr.scene.desc.camera
            for k in ("fov", "aperture", "focal_dist", "ortho"):
                v = getattr(req, k)
                if v is not None:
                    setattr(cam, k, v)
            viewer.pyr.scene._set_camera_native()
            viewer.pyr.scene.native.finalize()
            viewer.pyr.renderer.reset()
        return {"ok": True}

    @app.post("/api/camera/move")
    def cam_move(req: MoveReq):
        viewer.move_camera(req.key)
        return {"ok": True}

    return app


def main(argv=None):
    import argparse
    ap = argparse.ArgumentParser("hippt.viewer")
    ap.add_argument("scene", nargs="?", default="cornell")
    ap.add_argument("--device", type=int, default=None)
    ap.add_argument("--width", type=int, default=512)
    ap.add_argument("--height", type=int, default=512)
    ap.add_argument("--port", type=int, default=8517)
    args = ap.parse_args(argv)
    from ..scene import procedural
    if args.scene.endswith(".xml"):
        from ..scene.xml_parser import parse_xml
        desc = parse_xml(args.scene)
    else:
        desc = {"cornell": procedural.cornell_box, "kitchen": procedural.kitchen,
                "sports-car": procedural.sports_car, "smoke": procedural.smoke_box}[
            args.scene](width=args.width, height=args.height)
        desc.config.max_depth = max(desc.config.max_depth, 5)
    viewer = ViewerApp(desc, device=args.device)
    viewer.start()
    app = build_app(viewer)
    import uvicorn
    uvicorn.run(app, host="127.0.0.1", port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
