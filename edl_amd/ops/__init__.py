"""edl_amd.ops — the CDNA4 HIP kernel layer.

The extension `edl_amd._C` is built IN-TREE for gfx950 only
(`python setup.py build_ext --inplace`, driven by hipcc with
PYTORCH_ROCM_ARCH=gfx950; see csrc/). There is no CUDA path, no Triton,
no multi-backend dispatch: on a GPU box the HIP extension is REQUIRED —
a missing .so raises instead of silently falling back to eager torch
(set EDL_ALLOW_NO_EXT=1 to override for bring-up only). On CPU (tests)
torch reference implementations are used.
"""
import os

import torch

_EXT = None
_EXT_ERR = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from .. import _C  # built in-tree by setup.py build_ext --inplace

        _EXT = _C
    except ImportError as e:
        _EXT_ERR = e
    return _EXT


def available():
    return _try_load() is not None


def ext():
    """The extension module; raises loudly on a GPU box if missing."""
    m = _try_load()
    if m is None:
        if torch.cuda.is_available() and os.environ.get("EDL_ALLOW_NO_EXT") != "1":
            raise RuntimeError(
                "edl_amd._C HIP extension not built but a GPU is present. "
                "Build with `python setup.py build_ext --inplace` "
                "(PYTORCH_ROCM_ARCH=gfx950). Original error: %s" % _EXT_ERR
            )
        raise ImportError(str(_EXT_ERR))
    return m


def assert_hip_ops_available(model=None):
    """Fail LOUDLY if the HIP extension is missing on a GPU host.

    The fused modules (Conv2dFast, BNReLU2d, FusedSGD, avgpool, KD-CE)
    are baked into the model classes themselves — nothing is swapped at
    runtime; this is the engine's guard against silently training on
    torch fallbacks (the old name swap_module_ops
    implied a swap that never happened)."""
    if torch.cuda.is_available():
        ext()  # raises if the HIP extension is missing
    return model


# backward-compat alias (older scripts)
swap_module_ops = assert_hip_ops_available
