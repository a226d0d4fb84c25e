"""HIP extension loader.

The gfx950 extension is built IN-TREE (setup.py build_ext --inplace, or
__graft_entry__.build()) as ``real_time_helmet_detection_amd/ops/_C*.so`` so
the artifact travels with the repo snapshot to GPU boxes. There is no JIT
cache dependency and no fallback dispatch: on a CUDA(ROCm) tensor, a missing
extension is a hard error — the framework never silently runs eager torch on
the GPU (that would invalidate every benchmark and the native-code check).

``RTHD_EAGER_GPU=1`` exists ONLY for explicit A/B measurement of HIP kernels
vs torch-ROCm ops; it prints a warning once.
"""

import os
import sys

_ext = None
_tried = False
_err = None


def _load():
    global _ext, _tried, _err
    if _tried:
        return _ext
    _tried = True
    try:
        from . import _C  # built in-tree
        _ext = _C
    except ImportError as e:
        _err = e
        _ext = None
    return _ext


def ext():
    """The extension module, or None if not built."""
    return _load()


def require_ext():
    """The extension module; raises with build instructions if missing."""
    mod = _load()
    if mod is None:
        raise RuntimeError(
            'real_time_helmet_detection_amd HIP extension (_C) is not built '
            'for this tree — run `python setup.py build_ext --inplace` (or '
            '__graft_entry__.build()) with PYTORCH_ROCM_ARCH=gfx950. '
            f'Original import error: {_err}')
    return mod


_warned_eager = False


def eager_gpu_override():
    """True when RTHD_EAGER_GPU=1 explicitly requests torch-ROCm eager ops."""
    global _warned_eager
    on = os.environ.get('RTHD_EAGER_GPU', '0') == '1'
    if on and not _warned_eager:
        print('[rthd] WARNING: RTHD_EAGER_GPU=1 — running eager torch ops on '
              'GPU (A/B measurement mode, not the MI355X-native path)',
              file=sys.stderr)
        _warned_eager = True
    return on
