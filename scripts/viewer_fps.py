"""Measure viewer streaming fps at 1080p on one GPU (VERDICT r01 item 9
acceptance: >10 fps preview).  Times the full per-frame streaming path
(progressive render keeps accumulating in the background thread; each
frame_raw = GPU->host copy + tonemap + packet build)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    from hippt.scene.procedural import kitchen
    from hippt.viewer.server import ViewerApp
    desc = kitchen(width=1920, height=1080)
    v = ViewerApp(desc, device=0, spp_per_frame=1)
    v.start()
    try:
        time.sleep(1.0)          # warm: a few spp accumulated
        v.frame_raw()
        n = 40
        t0 = time.perf_counter()
        for _ in range(n):
            data = v.frame_raw()
        dt = time.perf_counter() - t0
        fps = n / dt
        print(f"[viewer-fps] 1080p stream: {fps:.1f} fps "
              f"({dt / n * 1000:.1f} ms/frame, packet {len(data) / 1e6:.1f} MB)")
        t0 = time.perf_counter()
        for _ in range(n):
            v.frame_raw(scale=2)
        dt = time.perf_counter() - t0
        print(f"[viewer-fps] 540p stream: {n / dt:.1f} fps")
    finally:
        v.stop()


if __name__ == "__main__":
    main()
