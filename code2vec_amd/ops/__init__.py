"""Compute ops for the code2vec_amd engine.

Two interchangeable backends behind one functional interface (`dispatch`):

- `reference` — pure PyTorch fp32/bf16 eager ops. The semantic oracle
  (matches the reference TF graph math, tensorflow_model.py:236-265,:226-230)
  and the CPU execution path.
- `hip` — hand-written CDNA4 (gfx950) HIP kernels, built in-tree as
  `code2vec_amd/ops/_c2v_hip*.so`. REQUIRED on GPU: if a CUDA/ROCm device is
  visible and the extension is missing, ops raise instead of silently falling
  back to eager (so a GPU run always exercises the native path).
"""

import os

import torch

_EXT = None
_EXT_ERR = None


def _try_load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    import importlib.util
    here = os.path.dirname(os.path.abspath(__file__))
    candidates = [os.path.join(here, '_c2v_hip.so'),
                  os.path.join(here, '_build', '_c2v_hip.so')]
    last_err = None
    for path in candidates:
        if not os.path.isfile(path):
            continue
        try:
            spec = importlib.util.spec_from_file_location('code2vec_amd.ops._c2v_hip', path)
            mod = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(mod)
            _EXT = mod
            return _EXT
        except Exception as e:  # noqa: BLE001
            last_err = e
    _EXT_ERR = last_err or FileNotFoundError(
        'no _c2v_hip.so found in %s' % (candidates,))
    return None


def hip_ext(required: bool = False):
    """Return the loaded HIP extension module (or None). With required=True,
    raise a loud error explaining how to build it."""
    ext = _try_load_extension()
    if ext is None and required:
        raise RuntimeError(
            "code2vec_amd HIP extension (_c2v_hip) is not built/importable, but a "
            "GPU device was requested. Build it in-tree with "
            "`python -m code2vec_amd.ops.build` (or `python __graft_entry__.py build`). "
            "Original import error: %r" % (_EXT_ERR,))
    return ext


def hip_available() -> bool:
    return _try_load_extension() is not None


def want_hip(tensor_or_device) -> bool:
    """True if this tensor/device should run on the HIP kernels."""
    if isinstance(tensor_or_device, torch.Tensor):
        return tensor_or_device.is_cuda
    return str(tensor_or_device).startswith('cuda')


_ALLOW_EAGER_GPU = os.environ.get('C2V_ALLOW_EAGER_GPU', '0') == '1'


def backend_for(tensor_or_device):
    """'hip' on GPU (extension mandatory unless C2V_ALLOW_EAGER_GPU=1), else 'ref'."""
    if want_hip(tensor_or_device):
        if hip_available():
            return 'hip'
        if _ALLOW_EAGER_GPU:
            return 'ref'
        hip_ext(required=True)
    return 'ref'
