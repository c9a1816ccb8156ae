"""vearch_amd — MI355X-native rebuild of Vearch's Gamma vector-search hot
path. The compute lives in libgamma.so (HIP/gfx950, built
from vearch_amd/csrc); this package is the host-side mirror of the
reference's engine SDK plus test/bench marshalling helpers."""
from . import fbsenc, merge, proto  # noqa: F401
from .engine import GammaEngine, clear_kill, set_kill  # noqa: F401
