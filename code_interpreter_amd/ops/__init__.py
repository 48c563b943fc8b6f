"""gfx950 HIP kernel library + numpy routing.

``_hipops`` (C extension, built in-tree by ops/build.py) provides the raw
device API; ``hipnp`` provides DeviceArray and the numpy patches. Both are
import-path-flat so sandbox children can load them standalone via
APP_OPS_DIR without pulling in the control plane.
"""

import os
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
if _OPS_DIR not in sys.path:
    sys.path.append(_OPS_DIR)


def load_hipops():
    """Import the _hipops extension, building it if missing."""
    try:
        import _hipops
    except ImportError:
        from code_interpreter_amd.ops.build import build

        build()
        import _hipops
    return _hipops


def load_hipnp():
    import hipnp

    return hipnp
