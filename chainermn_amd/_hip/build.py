"""In-tree build of chainermn_amd._C via hipcc (gfx950, no hipify).

Usage: python -m chainermn_amd._hip.build [--force]

Compiles every csrc/*.hip + bindings.cpp with hipcc
--offload-arch=gfx950 and links one _C.so next to the package so the
built artifact travels with the repo snapshot (no JIT cache).
"""

import hashlib
import os
import subprocess
import sys
import sysconfig

PKG = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(PKG, '_hip', 'csrc')
OUT = os.path.join(PKG, '_C.so')
STAMP = os.path.join(PKG, '_hip', '.build_stamp')


def _torch_paths():
    import torch
    tdir = os.path.dirname(torch.__file__)
    return (os.path.join(tdir, 'include'),
            os.path.join(tdir, 'include', 'torch', 'csrc', 'api', 'include'),
            os.path.join(tdir, 'lib'))


def _sources():
    srcs = [os.path.join(CSRC, f) for f in sorted(os.listdir(CSRC))
            if f.endswith(('.hip', '.cpp'))]
    return srcs


def _digest_inputs():
    # headers participate in the staleness digest (but not compilation)
    return _sources() + [os.path.join(CSRC, f)
                         for f in sorted(os.listdir(CSRC))
                         if f.endswith('.h')]


def _digest(srcs):
    h = hashlib.sha256()
    for s in srcs:
        with open(s, 'rb') as f:
            h.update(f.read())
    return h.hexdigest()


def build(force=False, verbose=True):
    import torch
    srcs = _sources()
    digest = _digest(_digest_inputs())
    if not force and os.path.exists(OUT) and os.path.exists(STAMP):
        with open(STAMP) as f:
            if f.read().strip() == digest:
                if verbose:
                    print('chainermn_amd._C up to date')
                return OUT
    inc1, inc2, libdir = _torch_paths()
    pyinc = sysconfig.get_paths()['include']
    abi = '1' if torch._C._GLIBCXX_USE_CXX11_ABI else '0'
    objdir = os.path.join(PKG, '_hip', 'build')
    os.makedirs(objdir, exist_ok=True)
    common = [
        '--offload-arch=gfx950', '-O3', '-std=c++17', '-fPIC',
        '-D__HIP_PLATFORM_AMD__=1', '-DUSE_ROCM=1', '-DHIPBLAS_V2',
        '-D__HIP_NO_HALF_OPERATORS__=1', '-D__HIP_NO_HALF_CONVERSIONS__=1',
        '-DTORCH_EXTENSION_NAME=_C', '-DTORCH_API_INCLUDE_EXTENSION_H',
        '-D_GLIBCXX_USE_CXX11_ABI=' + abi,
        '-I' + inc1, '-I' + inc2, '-I' + pyinc, '-I' + CSRC,
        '-Wno-unused-result',
    ]
    objs = []
    procs = []
    for s in srcs:
        obj = os.path.join(objdir,
                           os.path.basename(s).rsplit('.', 1)[0] + '.o')
        objs.append(obj)
        cmd = ['hipcc', '-c', s, '-o', obj] + common
        if verbose:
            print(' '.join(cmd[:4]), '...')
        procs.append((s, subprocess.Popen(cmd, stdout=subprocess.PIPE,
                                          stderr=subprocess.STDOUT)))
    failed = False
    for s, p in procs:
        out, _ = p.communicate()
        if p.returncode != 0:
            failed = True
            sys.stderr.write('FAILED: {}\n{}\n'.format(
                s, out.decode(errors='replace')))
        elif out.strip() and verbose:
            sys.stderr.write(out.decode(errors='replace'))
    if failed:
        raise RuntimeError('hipcc compilation failed')
    link = (['hipcc', '-shared', '-fPIC', '-o', OUT] + objs +
            ['-L' + libdir, '-ltorch', '-ltorch_cpu', '-ltorch_python',
             '-lc10', '-lc10_hip', '-ltorch_hip',
             '-Wl,-rpath,' + libdir,
             '-L/opt/rocm/lib', '-lamdhip64'])
    if verbose:
        print('linking', OUT)
    r = subprocess.run(link, capture_output=True)
    if r.returncode != 0:
        raise RuntimeError('link failed:\n' + r.stderr.decode())
    with open(STAMP, 'w') as f:
        f.write(digest)
    if verbose:
        print('built', OUT)
    return OUT


if __name__ == '__main__':
    build(force='--force' in sys.argv)
