"""Build entry: `python setup.py build_ext --inplace` compiles the gfx950
HIP extension in-tree (see dalle_pytorch_amd/ops/hip/build.py)."""

import sys

from setuptools import setup, find_packages

if 'build_ext' in sys.argv:
    from dalle_pytorch_amd.ops.hip.build import build
    build(force='--force' in sys.argv)
    sys.argv = [a for a in sys.argv if a not in ('build_ext', '--inplace', '--force')]
    if len(sys.argv) == 1:
        sys.exit(0)

setup(
    name='dalle-pytorch-amd',
    version='0.1.0',
    description='MI355X-native DALL-E training/generation framework',
    packages=find_packages(exclude=('tests',)),
    package_data={'dalle_pytorch_amd': ['data/*.txt.gz', 'ops/hip/*.hip']},
    include_package_data=True,
    python_requires='>=3.9',
)
