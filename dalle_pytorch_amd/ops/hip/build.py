"""In-tree hipcc build for the gfx950 extension.

Drives hipcc directly (no torch hipify pass — the sources are native HIP)
and drops ``dalle_pytorch_amd/_hip.so`` next to the package so the built
artifact travels with repo snapshots to GPU boxes. Cross-compiles fine on
machines without a GPU.
"""

import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parents[2]   # dalle_pytorch_amd/
SRC_DIR = Path(__file__).resolve().parent
SOURCES = [SRC_DIR / 'hip_ops.hip']
OUTPUT = PKG_DIR / '_hip.so'


def build(verbose=True, force=False):
    import torch
    from torch.utils import cpp_extension as ce

    if not force and OUTPUT.exists():
        newest_src = max(s.stat().st_mtime for s in SOURCES)
        if OUTPUT.stat().st_mtime > newest_src:
            if verbose:
                print(f'[build] {OUTPUT.name} up to date')
            return str(OUTPUT)

    hipcc = str(Path(ce.ROCM_HOME or '/opt/rocm') / 'bin' / 'hipcc')
    abi = '1' if torch._C._GLIBCXX_USE_CXX11_ABI else '0'
    inc = [f'-I{p}' for p in ce.include_paths()]
    inc.append(f"-I{sysconfig.get_paths()['include']}")
    libdirs = [f'-L{p}' for p in ce.library_paths()]

    cmd = [
        hipcc, '-O3', '-std=c++17', '--offload-arch=gfx950',
        '-fPIC', '-shared',
        '-DTORCH_EXTENSION_NAME=_hip',
        '-DTORCH_API_INCLUDE_EXTENSION_H',
        f'-D_GLIBCXX_USE_CXX11_ABI={abi}',
        *ce.COMMON_HIP_FLAGS, *ce.COMMON_HIPCC_FLAGS,
        '-DTORCH_HIP_VERSION=' + torch.version.hip.split('.')[0],
        '-Wno-deprecated-declarations', '-Wno-unused-result',
        *inc, *[str(s) for s in SOURCES], *libdirs,
        '-ltorch', '-ltorch_cpu', '-ltorch_hip', '-lc10', '-lc10_hip',
        '-ltorch_python', '-lamdhip64',
        '-o', str(OUTPUT),
    ]
    if verbose:
        print('[build]', ' '.join(cmd))
    subprocess.run(cmd, check=True)
    return str(OUTPUT)


if __name__ == '__main__':
    build(force='--force' in sys.argv)
