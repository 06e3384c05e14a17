"""In-tree build of the parallax_amd native extension for gfx950.

Usage: ``python -m parallax_amd.ops.build`` (no GPU required — hipcc
cross-compiles).  The resulting ``parallax_amd_C`` shared object lands in
``parallax_amd/_lib/`` inside the repo (NOT a cache dir), so the binary
travels with the source tree to the GPU box.
"""

import glob
import os

PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB_DIR = os.path.join(PKG_DIR, '_lib')
CSRC = os.path.join(PKG_DIR, 'ops', 'hip', 'csrc')
EXT_NAME = 'parallax_amd_C'


def build(verbose=True):
    os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    os.environ.setdefault('MAX_JOBS', str(min(os.cpu_count() or 8, 16)))
    from torch.utils import cpp_extension

    os.makedirs(LIB_DIR, exist_ok=True)
    sources = sorted(
        glob.glob(os.path.join(CSRC, '*.cpp')) +
        [s for s in glob.glob(os.path.join(CSRC, '*.hip'))
         if not s.endswith('_hip.hip')])  # torch-hipify build artifacts
    assert sources, 'no kernel sources found under %s' % CSRC
    mod = cpp_extension.load(
        name=EXT_NAME,
        sources=sources,
        extra_include_paths=[CSRC],
        extra_cflags=['-O3'],
        extra_cuda_cflags=['-O3'],
        build_directory=LIB_DIR,
        verbose=verbose,
        with_cuda=True,
    )
    return mod


if __name__ == '__main__':
    build()
    print('built %s into %s' % (EXT_NAME, LIB_DIR))
