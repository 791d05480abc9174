"""In-tree build of the CDNA4 HIP extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands inside adaqp_amd/ so it travels with the repo snapshot to
GPU boxes (no JIT cache dependence).
"""
import os

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name='adaqp_amd',
    version='0.1.0',
    packages=['adaqp_amd'],
    ext_modules=[
        CUDAExtension(
            name='adaqp_amd._C',
            sources=['adaqp_amd/csrc/kernels.hip'],
            extra_compile_args={'cxx': ['-O3'],
                                'nvcc': ['-O3', '--offload-arch=gfx950']},
        )
    ],
    cmdclass={'build_ext': BuildExtension},
)
