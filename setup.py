"""Build the sat_amd._C HIP extension in-tree (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built sat_amd/_C*.so lives next to the package so it ships with the
repo snapshot to GPU boxes (it is git-ignored; history stays source-only).
"""

import os

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, 'sat_amd', 'ops', 'csrc')

setup(
    name='sat_amd',
    version='0.1.0',
    description='MI355X-native Show-Attend-and-Tell framework',
    packages=['sat_amd'],
    ext_modules=[
        CUDAExtension(
            name='sat_amd._C',
            sources=[
                os.path.join(CSRC, 'bindings.cpp'),
                os.path.join(CSRC, 'gemm.hip'),
                os.path.join(CSRC, 'kernels.hip'),
                os.path.join(CSRC, 'bptt_fuse.hip'),
                os.path.join(CSRC, 'conv3.hip'),
                os.path.join(CSRC, 'conv8p.hip'),
                os.path.join(CSRC, 'conv_bwd.hip'),
                os.path.join(CSRC, 'gemm8p.hip'),
            ],
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17'],
            },
        ),
    ],
    cmdclass={'build_ext': BuildExtension},
)
