"""hippt — MI355X-native software path tracing framework.

A from-scratch re-design of the capability set of the reference CUDA renderer
(Enigmatisms/cuda-pt) for AMD Instinct MI355X (gfx950, CDNA4): megakernel /
wavefront / volumetric path tracing, light tracing, SAH-BVH + SBVH, an 8-type
BSDF system, Mitsuba-like XML scenes, a PythonRenderer returning PyTorch-ROCm
tensors, and multi-GPU sample-split DDP rendering over RCCL/xGMI.
"""

__version__ = "0.2.0"

import os


def _load_native():
    # torch MUST be imported before the native extension: torch bundles its own
    # ROCm runtime (torch/lib/libamdhip64.so, soname libamdhip64.so.7).  When
    # torch loads first, _C's libamdhip64.so.7 dependency resolves to torch's
    # already-loaded copy and both share ONE HIP/HSA runtime; the other order
    # creates two HSA instances and hipSetDevice fails with hipErrorNoDevice.
    try:
        import torch  # noqa: F401
    except ImportError:
        pass
    try:
        from . import _C  # noqa: F401
        return _C
    except ImportError as e:
        raise ImportError(
            "hippt._C native extension is not built. Run "
            "`python setup.py build_ext --inplace` at the repo root "
            f"(original error: {e})"
        ) from e


C = _load_native()

# On a machine with a visible GPU the native HIP path is mandatory: fail loudly
# rather than silently falling back to CPU rendering.
def has_gpu() -> bool:
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


from .pyrender import PythonRenderer  # noqa: E402,F401
from .scene.scene import Scene, SceneDesc  # noqa: E402,F401
from .render.renderer import Renderer  # noqa: E402,F401
