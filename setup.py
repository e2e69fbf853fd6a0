"""In-tree build of the MI355X (gfx950) HIP extensions.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces se3_transformer_amd/_C*.so next to the package sources so the
snapshot shipped to GPU boxes carries the built extension.
"""
import os

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from setuptools import find_packages, setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

setup(
    name='se3_transformer_amd',
    version='0.9.0',
    description='MI355X-native SE(3)-equivariant transformer '
                '(API-compatible with se3-transformer-pytorch 0.9.0)',
    packages=find_packages(include=['se3_transformer_amd*']),
    python_requires='>=3.9',
    install_requires=['torch>=2.0', 'numpy'],
    ext_modules=[
        CUDAExtension(
            name='se3_transformer_amd._C',
            sources=['se3_transformer_amd/csrc/pairconv.hip',
                     'se3_transformer_amd/csrc/pairconv_bwd.hip',
                     'se3_transformer_amd/csrc/pack_w.hip',
                     'se3_transformer_amd/csrc/radial.hip',
                     'se3_transformer_amd/csrc/ubuild.hip',
                     'se3_transformer_amd/csrc/egnn.hip',
                     'se3_transformer_amd/csrc/sh_basis.hip',
                     'se3_transformer_amd/csrc/norm_se3.hip',
                     'se3_transformer_amd/csrc/attn2.hip',
                     'se3_transformer_amd/csrc/knn.hip'],
            extra_compile_args={
                'cxx': ['-O3'],
                'nvcc': ['-O3', '--offload-arch=gfx950'] +
                        (['-Rpass-analysis=kernel-resource-usage']
                         if os.environ.get('SE3_RES_USAGE') else []),
            },
        ),
    ],
    cmdclass={'build_ext': BuildExtension.with_options(no_python_abi_suffix=False)},
)
