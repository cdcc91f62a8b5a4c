"""Native-kernel dispatch policy.

On a GPU box the hand-written CDNA4 kernels (easyparallellibrary_amd._C)
are the compute path; a missing extension there is a hard error — no
silent eager fallback.  On CPU the pure-torch reference paths run (and are
what the numerics tests compare the HIP kernels against).
"""

import torch

_EXT = None
_EXT_ERR = None


def native_ext():
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from easyparallellibrary_amd import _C
            _EXT = _C
        except ImportError as e:  # pragma: no cover
            _EXT_ERR = e
    if _EXT is None:
        raise RuntimeError(
            "easyparallellibrary_amd._C native extension is not built "
            "(required on GPU; build with "
            "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace): "
            "{}".format(_EXT_ERR))
    return _EXT


def native_available():
    try:
        native_ext()
        return True
    except RuntimeError:
        return False


def use_native(t):
    """Native kernels run for device tensors.  If the tensor is on GPU and
    the extension is missing, raise (fail loudly on a GPU box)."""
    if not t.is_cuda:
        return False
    from easyparallellibrary_amd.env import Env
    if not Env.get().config.kernel.fused:
        return False
    native_ext()  # raises if missing
    return True
