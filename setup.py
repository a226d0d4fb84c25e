"""In-tree build of the gfx950 HIP extension.

PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces real_time_helmet_detection_amd/ops/_C*.so next to its python
wrappers so the artifact travels with repo snapshots (no JIT cache).
"""

import os
import glob

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, 'real_time_helmet_detection_amd', 'ops', 'csrc')

# exclude hipify build artifacts (*_hip.hip) that torch's build may drop
# next to the real sources — they are gitignored copies, not sources
sources = sorted(p for p in
                 glob.glob(os.path.join(CSRC, '*.cpp')) +
                 glob.glob(os.path.join(CSRC, '*.hip'))
                 if not p.endswith('_hip.hip'))

setup(
    name='real_time_helmet_detection_amd',
    version='0.1.0',
    packages=['real_time_helmet_detection_amd'],
    ext_modules=[
        CUDAExtension(
            name='real_time_helmet_detection_amd.ops._C',
            sources=sources,
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17'],
            },
        )
    ],
    cmdclass={'build_ext': BuildExtension.with_options(use_ninja=True)},
)
