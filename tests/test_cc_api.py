"""C++ API (Scope/ClientSession/AddSymbolicGradients): compile
tests/c/cc_api_test.cc against libstf_c.so and run it."""
import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLIB = os.path.join(REPO, 'simple_tensorflow_amd', 'libstf_c.so')


@pytest.mark.skipif(not os.path.exists(CLIB), reason='libstf_c.so not built')
def test_cc_api_smoke(tmp_path):
    exe = str(tmp_path / 'cc_api_test')
    csrc = os.path.join(REPO, 'simple_tensorflow_amd', 'csrc')
    subprocess.run(
        ['g++', '-std=c++17', '-o', exe,
         os.path.join(REPO, 'tests', 'c', 'cc_api_test.cc'),
         '-I', csrc, '-D__HIP_PLATFORM_AMD__', '-I/opt/rocm/include',
         '-L', os.path.dirname(CLIB), '-lstf_c',
         '-Wl,-rpath,' + os.path.dirname(CLIB),
         '-Wl,-rpath,/opt/rocm/lib'],
        check=True, capture_output=True)
    r = subprocess.run([exe], capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr + r.stdout
    assert 'CC_API_OK' in r.stdout
