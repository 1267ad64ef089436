"""Golden-image regression: bit-level CPU renders of the example scenes are
compared against committed references (tests/golden/*.npz, regenerated with
scripts/make_goldens.py after INTENTIONAL semantics changes).  Catches
subtle BSDF/integrator/sampler regressions that statistical tests miss."""
import os

import numpy as np
import pytest

import hippt
from hippt.scene.xml_parser import parse_xml

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GOLD = os.path.join(ROOT, "tests", "golden")

NAMES = ["cornell-box", "balls", "grid-cbox", "diamonds", "env-balls",
         "point-cbox", "hero", "grid-cbox-nvdb", "medium-cbox", "water-cbox"]
PROC = ["kitchen", "sports-car"]


@pytest.mark.parametrize("name", NAMES)
def test_golden(name):
    z = np.load(os.path.join(GOLD, name + ".npz"))
    ref, spp, w = z["img"], int(z["spp"]), int(z["w"])
    d = parse_xml(os.path.join(ROOT, "scenes", name + ".xml"))
    d.camera.height = max(16, int(w * d.camera.height / d.camera.width))
    d.camera.width = w
    img = hippt.PythonRenderer(d, device_id=-1).render(spp=spp).numpy()
    assert img.shape == ref.shape
    # same seeds + deterministic CPU path -> near-bit-exact; small rtol for
    # cross-compiler fp differences
    np.testing.assert_allclose(img, ref, rtol=2e-4, atol=2e-4, err_msg=name)


@pytest.mark.parametrize("name", PROC)
def test_golden_procedural(name):
    from hippt.scene.procedural import kitchen, sports_car
    z = np.load(os.path.join(GOLD, name + ".npz"))
    gen = {"kitchen": kitchen, "sports-car": sports_car}[name]
    d = gen(width=64, height=36)
    img = hippt.PythonRenderer(d, device_id=-1).render(spp=int(z["spp"])).numpy()
    np.testing.assert_allclose(img, z["img"], rtol=2e-4, atol=2e-4, err_msg=name)
