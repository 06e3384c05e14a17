"""pip packaging (reference parallax/util/setup.py + build_pip_package.sh).

Builds the gfx950 HIP extension in-tree first:
    PYTORCH_ROCM_ARCH=gfx950 python -m parallax_amd.ops.build
    python setup.py bdist_wheel
"""

import os

from setuptools import setup, find_packages

HERE = os.path.dirname(os.path.abspath(__file__))

setup(
    name='parallax_amd',
    version='0.1.0',
    description=('MI355X-native sparsity-aware data-parallel training '
                 'framework (capabilities of snuspl/parallax, re-designed '
                 'for CDNA4 + RCCL over xGMI)'),
    packages=find_packages(include=['parallax_amd', 'parallax_amd.*']),
    package_data={'parallax_amd': ['_lib/*.so', 'ops/hip/csrc/*']},
    python_requires='>=3.8',
    install_requires=['numpy', 'scipy'],
)
