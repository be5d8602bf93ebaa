"""Build the in-tree HIP extension for gfx950.

Compiles handyrl_amd/ops/src/*.hip with hipcc (via torch.utils.cpp_extension)
into handyrl_amd/ops/_C.so.  hipcc cross-compiles without a GPU, so this runs
on CPU-only build hosts; the .so travels with the repo snapshot.
"""

import os
import shutil
import sys


def build(verbose=False):
    os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    os.environ.setdefault('MAX_JOBS', '8')
    from torch.utils import cpp_extension

    here = os.path.dirname(os.path.abspath(__file__))
    src_dir = os.path.join(here, 'src')
    build_dir = os.path.join(here, '_build')
    os.makedirs(build_dir, exist_ok=True)

    # List sources explicitly: torch's hipify pass writes generated
    # `*_hip.hip` siblings next to the source, which a bare glob would pick
    # up on the next build (duplicate kernels / double op registration).
    sources = [os.path.join(src_dir, 'ext.hip')]

    mod = cpp_extension.load(
        name='handyrl_amd_C',
        sources=sources,
        build_directory=build_dir,
        extra_cflags=['-O3', '-std=c++17'],
        extra_cuda_cflags=['-O3', '-std=c++17'],
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )
    so = os.path.join(build_dir, 'handyrl_amd_C.so')
    if not os.path.exists(so):
        cands = [f for f in os.listdir(build_dir) if f.endswith('.so')]
        if not cands:
            raise RuntimeError('extension build produced no .so in %s' % build_dir)
        so = os.path.join(build_dir, cands[0])
    dst = os.path.join(here, '_C.so')
    shutil.copy2(so, dst)
    print('built %s' % dst)
    return dst


if __name__ == '__main__':
    build(verbose='-v' in sys.argv)
