"""Build the in-tree CPU env-core extension (g++ + pybind11).

handyrl_amd/envs/src/vec_geese_core.cpp -> handyrl_amd/envs/_vec_geese_core.so
— a host-side native module (no GPU, no torch linkage): the env workers'
hot loop.  The .so travels with repo snapshots like the HIP extension.
"""

import os
import subprocess
import sys
import sysconfig


def build(verbose=False):
    import pybind11
    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, 'src', 'vec_geese_core.cpp')
    out = os.path.join(here, '_vec_geese_core.so')
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


if __name__ == '__main__':
    build(verbose='-v' in sys.argv)
