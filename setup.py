"""Build the in-tree gfx950 HIP extension: python setup.py build_ext --inplace.

The .so lands at timm_amd/_C*.so (git-ignored; ships to GPU boxes with the
snapshot). hipcc cross-compiles without a GPU present.
"""
import os
import sys

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

CSRC = os.path.join('timm_amd', 'ops', 'csrc')

sources = [
    os.path.join(CSRC, 'bindings.cpp'),
    os.path.join(CSRC, 'layernorm.hip'),
    os.path.join(CSRC, 'elementwise.hip'),
    os.path.join(CSRC, 'attention.hip'),
    os.path.join(CSRC, 'attention_bwd.hip'),
    os.path.join(CSRC, 'depthwise_conv.hip'),
    os.path.join(CSRC, 'multi_tensor.hip'),
    os.path.join(CSRC, 'muon_ns.hip'),
    os.path.join(CSRC, 'ce_loss.hip'),
    os.path.join(CSRC, 'data_ops.hip'),
]

setup(
    name='timm_amd_ext',
    ext_modules=[
        CUDAExtension(
            name='timm_amd._C',
            sources=sources,
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17', '--offload-arch=gfx950'],
            },
        ),
    ],
    cmdclass={'build_ext': BuildExtension.with_options(no_python_abi_suffix=False)},
)
