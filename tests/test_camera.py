"""Camera-model tests: thin-lens DoF, orthographic projection, hflip
(reference camera_model.cuh:58-104 features)."""
import numpy as np

import hippt
from hippt.scene.scene import (SceneDesc, ObjectDesc, BsdfDesc, EmitterDesc,
                               CameraDesc, RenderConfig)


def two_spheres(aperture=0.0, focal_dist=0.0, ortho=False, hflip=False):
    d = SceneDesc()
    d.bsdfs = [BsdfDesc(type="lambertian", kd=(0.8, 0.2, 0.2)),
               BsdfDesc(type="lambertian", kd=(0.2, 0.2, 0.8))]
    d.emitters = [EmitterDesc(type="point", pos=(0, 6, -4), emission=(1, 1, 1),
                              scale=60.0)]
    d.objects = [ObjectDesc(spheres=np.array([[-0.9, 0, 0, 0.5]], np.float32), bsdf=0),
                 ObjectDesc(spheres=np.array([[0.9, 0, 6, 0.5]], np.float32), bsdf=1)]
    d.camera = CameraDesc(pos=(0, 0, -4), lookat=(0, 0, 1), fov=30,
                          width=96, height=96, aperture=aperture,
                          focal_dist=focal_dist, ortho=ortho, hflip=hflip)
    d.config = RenderConfig(renderer="pt", spp=1, max_depth=2)
    return d


def edge_sharpness(img, col_range):
    """Max horizontal gradient within a column range (sharp edge = large)."""
    g = np.abs(np.diff(img[..., :3].sum(axis=2), axis=1))
    return float(g[:, col_range].max())


def test_depth_of_field():
    sharp = hippt.PythonRenderer(two_spheres(), device_id=-1).render(spp=48).numpy()
    dof = hippt.PythonRenderer(two_spheres(aperture=0.35, focal_dist=4.0),
                               device_id=-1).render(spp=48).numpy()
    # near sphere (left half) is at the focal plane: stays sharp-ish;
    # far sphere (right half) defocuses: its edge gradient collapses
    near_cols = np.s_[5:45]
    far_cols = np.s_[50:90]
    assert edge_sharpness(dof, far_cols) < 0.5 * edge_sharpness(sharp, far_cols)
    assert edge_sharpness(dof, near_cols) > 0.5 * edge_sharpness(sharp, near_cols)


def test_orthographic_equal_size():
    d = two_spheres(ortho=True)
    d.camera.ortho_scale = 3.0 / 96
    img = hippt.PythonRenderer(d, device_id=-1).render(spp=24).numpy()
    red = (img[..., 0] > 2 * img[..., 2]) & (img[..., :3].sum(axis=2) > 0.05)
    blue = (img[..., 2] > 2 * img[..., 0]) & (img[..., :3].sum(axis=2) > 0.05)
    # parallel projection: same radius -> same pixel area despite 6 units of
    # depth difference (perspective would shrink the far one dramatically)
    assert red.sum() > 50 and blue.sum() > 50
    assert 0.7 < blue.sum() / red.sum() < 1.4, (red.sum(), blue.sum())


def test_hflip_mirrors_image():
    a = hippt.PythonRenderer(two_spheres(), device_id=-1).render(spp=32).numpy()
    b = hippt.PythonRenderer(two_spheres(hflip=True), device_id=-1).render(spp=32).numpy()
    flipped = b[:, ::-1]
    # same scene mirrored; compare red-sphere mass per half
    left_a = a[:, :48, 0].sum()
    left_f = flipped[:, :48, 0].sum()
    assert abs(left_a - left_f) < 0.1 * max(left_a, 1e-9)
    assert abs(a[..., :3].mean() - b[..., :3].mean()) < 0.05 * a[..., :3].mean()
