"""Compute-op dispatch.

CPU tensors run the PyTorch reference implementations (:mod:`.cpu`).
CUDA (ROCm) tensors run the hand-written CDNA4 HIP kernels (:mod:`.gpu`),
and FAIL LOUDLY if the compiled extension is missing — there is no silent
eager fallback on a GPU box (a fallback would invisibly bench the wrong
code path).
"""

from __future__ import annotations

import torch

from . import cpu as _cpu

_gpu = None
_gpu_err: Exception | None = None


def _gpu_mod():
    global _gpu, _gpu_err
    if _gpu is None and _gpu_err is None:
        try:
            from . import gpu as gpu_mod

            _gpu = gpu_mod
        except Exception as e:  # pragma: no cover - exercised on GPU boxes
            _gpu_err = e
    if _gpu is None:
        raise RuntimeError(
            "CUDA tensor passed to traffic_classifier_sdn_amd.ops but the HIP "
            "extension (_tcsdn_hip) is not available. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_gpu_err!r}"
        )
    return _gpu


def _mod_for(t: torch.Tensor):
    return _gpu_mod() if t.is_cuda else _cpu


def __getattr__(name):
    # expose every op from the cpu module's public surface; dispatch happens
    # inside the wrapper at call time based on the first tensor argument.
    if name.startswith("_"):
        raise AttributeError(name)
    fn_cpu = getattr(_cpu, name, None)
    if fn_cpu is None:
        raise AttributeError(name)

    def dispatcher(*args, **kwargs):
        first = next((a for a in args if isinstance(a, torch.Tensor)), None)
        mod = _mod_for(first) if first is not None else _cpu
        fn = getattr(mod, name, None)
        if fn is None:
            # The gpu module must bind every op explicitly (it may alias a
            # cold op to the torch impl on purpose); anything else is a
            # missing-kernel bug, not a fallback.
            raise RuntimeError(
                f"op {name!r} has no GPU implementation bound in ops.gpu"
            )
        return fn(*args, **kwargs)

    dispatcher.__name__ = name
    dispatcher.__doc__ = fn_cpu.__doc__
    return dispatcher
