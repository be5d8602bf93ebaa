"""Build the in-tree CPU env-core extensions (g++ + pybind11).

handyrl_amd/envs/src/vec_geese_core.cpp  -> envs/_vec_geese_core.so
handyrl_amd/envs/src/vec_geister_core.cpp -> envs/_vec_geister_core.so
— host-side native modules (no GPU, no torch linkage): the env workers'
hot loops.  The .so files travel with repo snapshots like the HIP
extension.
"""

import os
import subprocess
import sys
import sysconfig

CORES = ['vec_geese_core', 'vec_geister_core']


def _build_one(name, verbose=False):
    import pybind11
    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, 'src', name + '.cpp')
    out = os.path.join(here, '_%s.so' % name)
    if os.path.exists(out) and os.path.getmtime(out) > os.path.getmtime(src):
        return out
    cmd = [
        'g++', '-O3', '-std=c++17', '-shared', '-fPIC',
        '-fvisibility=hidden',
        '-I' + pybind11.get_include(),
        '-I' + sysconfig.get_paths()['include'],
        src, '-o', out,
    ]
    if verbose:
        print(' '.join(cmd))
    subprocess.run(cmd, check=True)
    print('built %s' % out)
    return out


def build(verbose=False):
    return [_build_one(name, verbose) for name in CORES]


if __name__ == '__main__':
    build(verbose='-v' in sys.argv)
