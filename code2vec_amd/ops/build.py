"""In-tree build of the CDNA4 HIP extension.

Builds `code2vec_amd/ops/_build/_c2v_hip.so` for gfx950 via
torch.utils.cpp_extension (which drives hipcc for the .hip source) and copies
it next to this file so the loader (ops/__init__.py) and the gpurun snapshot
both see it. hipcc cross-compiles without a GPU, so this runs on the CPU-only
build host too.

Usage: python -m code2vec_amd.ops.build
"""

import os
import shutil
import sys


def _is_fresh(dest: str, *sources: str) -> bool:
    """True if dest exists and is newer than every source. Used to avoid
    re-invoking cpp_extension.load in a process that may already have the
    module loaded (double pybind init in one process segfaults)."""
    if not os.path.isfile(dest):
        return False
    dest_m = os.path.getmtime(dest)
    return all(os.path.getmtime(s) < dest_m for s in sources if os.path.isfile(s))


def build(verbose: bool = True) -> str:
    os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    here = os.path.dirname(os.path.abspath(__file__))
    build_dir = os.path.join(here, '_build')
    os.makedirs(build_dir, exist_ok=True)
    src = os.path.join(here, 'csrc', 'c2v_kernels.hip')
    dest = os.path.join(here, '_c2v_hip.so')
    if _is_fresh(dest, src):
        build_reader(verbose=verbose)
        return dest

    from torch.utils import cpp_extension
    cpp_extension.load(
        name='_c2v_hip',
        sources=[src],
        build_directory=build_dir,
        extra_cflags=['-O3'],
        # AGPR-form MFMA: accumulators allocate in the AGPR half of the
        # unified gfx950 register file — without this the 192-reg split-K
        # accumulator tile spills (25 in-loop scratch ops)
        extra_cuda_cflags=['-O3', '-std=c++17',
                           '-mllvm', '-amdgpu-mfma-vgpr-form=0'],
        verbose=verbose,
        is_python_module=False,
        with_cuda=True,
    )
    built = os.path.join(build_dir, '_c2v_hip.so')
    shutil.copy2(built, dest)
    build_reader(verbose=verbose)
    return dest


def build_asan(verbose: bool = True) -> str:
    """Device-ASAN instrumented build of the kernel extension
    (gfx950:xnack+): the GPU-side sanitizer pass. Loaded only by
    tests/test_gpu_sanitizer.py in a subprocess with HSA_XNACK=1 and the
    host ASAN runtime LD_PRELOADed."""
    os.environ['PYTORCH_ROCM_ARCH'] = 'gfx950:xnack+'
    here = os.path.dirname(os.path.abspath(__file__))
    build_dir = os.path.join(here, '_build_asan')
    os.makedirs(build_dir, exist_ok=True)
    src = os.path.join(here, 'csrc', 'c2v_kernels.hip')
    dest = os.path.join(here, '_c2v_hip_asan.so')
    try:
        if _is_fresh(dest, src):
            return dest
        # The build runs in a SACRIFICIAL subprocess: cpp_extension.load
        # dlopens the result after building, and loading an ASAN .so into a
        # non-ASAN python _exit()s the process ("ASan runtime does not come
        # first"). The ninja build completes before that, so the artifact
        # survives; consumers LD_PRELOAD the runtime.
        import subprocess
        import sys
        code = (
            "from torch.utils import cpp_extension\n"
            "cpp_extension.load(name='_c2v_hip_asan', sources=[%r],\n"
            "    build_directory=%r,\n"
            "    extra_cflags=['-O1', '-g'],\n"
            "    extra_cuda_cflags=['-O1', '-g', '-std=c++17',\n"
            "                       '-fsanitize=address', '-shared-libsan'],\n"
            "    extra_ldflags=['-L/opt/rocm/lib/llvm/lib/clang/22/lib/linux',\n"
            "                   '-lclang_rt.asan-x86_64',\n"
            "                   '-Wl,-rpath,/opt/rocm/lib/llvm/lib/clang/22/lib/linux'],\n"
            "    verbose=%r, is_python_module=False, with_cuda=True)\n"
        ) % (src, build_dir, verbose)
        env = dict(os.environ, PYTORCH_ROCM_ARCH='gfx950:xnack+')
        subprocess.run([sys.executable, '-c', code], env=env,
                       capture_output=not verbose)
        built = os.path.join(build_dir, '_c2v_hip_asan.so')
        if not os.path.isfile(built):
            raise RuntimeError('ASAN extension build failed')
        shutil.copy2(built, dest)
    finally:
        os.environ['PYTORCH_ROCM_ARCH'] = 'gfx950'
    return dest


def build_reader(verbose: bool = True) -> str:
    """CPU-only native reader extension (no HIP): built into
    code2vec_amd/data/_c2v_reader.so."""
    here = os.path.dirname(os.path.abspath(__file__))
    data_dir = os.path.join(os.path.dirname(here), 'data')
    build_dir = os.path.join(data_dir, '_build')
    os.makedirs(build_dir, exist_ok=True)
    src = os.path.join(data_dir, 'csrc', 'c2v_reader.cpp')
    dest = os.path.join(data_dir, '_c2v_reader.so')
    if _is_fresh(dest, src):
        return dest
    from torch.utils import cpp_extension
    cpp_extension.load(
        name='_c2v_reader',
        sources=[src],
        build_directory=build_dir,
        extra_cflags=['-O3', '-std=c++17'],
        verbose=verbose,
        is_python_module=False,
        with_cuda=False,
    )
    built = os.path.join(build_dir, '_c2v_reader.so')
    shutil.copy2(built, dest)
    return dest


if __name__ == '__main__':
    path = build()
    print('built:', path)
    sys.exit(0)
