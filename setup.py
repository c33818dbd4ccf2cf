"""Build for dmlcloud_amd: MI355X-native distributed training framework.

Builds the gfx950 HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting dmlcloud_amd/_C*.so travels with the source tree.
"""

import os

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from setuptools import find_packages, setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

CSRC = os.path.join('dmlcloud_amd', 'ops', 'csrc')

ext = CUDAExtension(
    name='dmlcloud_amd._C',
    sources=[
        os.path.join(CSRC, 'bindings.cpp'),
        os.path.join(CSRC, 'reduce.hip'),
        os.path.join(CSRC, 'copy.hip'),
        os.path.join(CSRC, 'optim.hip'),
        os.path.join(CSRC, 'smallcnn.hip'),
        os.path.join(CSRC, 'layernorm.hip'),
        os.path.join(CSRC, 'loss.hip'),
        os.path.join(CSRC, 'attention.hip'),
    ],
    extra_compile_args={
        'cxx': ['-O3', '-std=c++17'],
        'nvcc': ['-O3', '-std=c++17', '--offload-arch=gfx950'],
    },
)

setup(
    name='dmlcloud_amd',
    version='0.1.0',
    description='MI355X-native distributed training pipeline framework',
    packages=find_packages(include=['dmlcloud_amd', 'dmlcloud_amd.*']),
    ext_modules=[ext],
    cmdclass={'build_ext': BuildExtension},
    python_requires='>=3.10',
)
