"""In-tree build of the _kfaccore HIP extension for gfx950.

Drives hipcc directly (no hipify, no nvcc shims): the device sources in
csrc/ are native HIP/CDNA4. The resulting kfac_amd/_kfaccore.so lives
in-tree so it travels with repo snapshots.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(REPO_ROOT, 'csrc')
OUT = os.path.join(REPO_ROOT, 'kfac_amd', '_kfaccore.so')

SOURCES = [
    os.path.join(CSRC, 'syrk.hip'),
    os.path.join(CSRC, 'gemm.hip'),
    os.path.join(CSRC, 'eigh.hip'),
    os.path.join(CSRC, 'chol.hip'),
    os.path.join(CSRC, 'binding.cpp'),
]
HEADERS = [os.path.join(CSRC, 'common.h')]


def _needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    out_mtime = os.path.getmtime(OUT)
    return any(
        os.path.getmtime(src) > out_mtime for src in SOURCES + HEADERS
    )


def build(force: bool = False, arch: str = 'gfx950', verbose: bool = True) -> str:
    """Compile csrc/ into kfac_amd/_kfaccore.so with hipcc.

    Cross-compiles fine on a GPU-less box (hipcc needs no device).
    Returns the path to the built library.
    """
    if not force and not _needs_build():
        return OUT

    import torch
    import torch.utils.cpp_extension as ce

    includes = ce.include_paths('cuda') + [
        CSRC,
        sysconfig.get_paths()['include'],
    ]
    lib_dirs = ce.library_paths('cuda')
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)

    cmd = [
        'hipcc',
        f'--offload-arch={arch}',
        '-O3',
        '-std=c++17',
        '-fPIC',
        '-shared',
        *SOURCES,
        *[f'-I{p}' for p in includes],
        '-D__HIP_PLATFORM_AMD__=1',
        '-DUSE_ROCM=1',
        '-DHIPBLAS_V2',
        '-DHIP_ENABLE_WARP_SYNC_BUILTINS=1',
        '-DTORCH_EXTENSION_NAME=_kfaccore',
        '-DTORCH_API_INCLUDE_EXTENSION_H',
        f'-D_GLIBCXX_USE_CXX11_ABI={abi}',
        *[f'-L{p}' for p in lib_dirs],
        '-ltorch',
        '-ltorch_hip',
        '-lrocsolver',
        '-lrocblas',
        '-lc10',
        '-lc10_hip',
        '-ltorch_python',
        '-lamdhip64',
        *[f'-Wl,-rpath,{p}' for p in lib_dirs],
        '-o',
        OUT,
    ]
    if verbose:
        print('[kfac_amd build]', ' '.join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == '__main__':
    build(force='--force' in sys.argv)
