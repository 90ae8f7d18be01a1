"""Build the gfx950 HIP extension in-tree:
    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The built .so lands in heterofl_amd/ops/ and travels with the repo snapshot.
"""
import os

from setuptools import setup
from torch.utils import cpp_extension

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, 'heterofl_amd', 'ops', 'csrc')

sources = [os.path.join(CSRC, f) for f in sorted(os.listdir(CSRC))
           if f.endswith(('.cpp', '.hip'))
           and not f.endswith('_hip.hip')]  # hipify-pass copies of our .hip

setup(
    name='heterofl_amd_hip',
    ext_modules=[
        cpp_extension.CUDAExtension(
            name='heterofl_amd.ops._heterofl_hip',
            sources=sources,
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17'],
            },
        )
    ],
    cmdclass={'build_ext': cpp_extension.BuildExtension},
)
