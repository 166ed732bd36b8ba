"""Native-op dispatch for cpd_amd.

Every op has a CPU implementation (``_cpd_cpu``, g++) and a gfx950 HIP
implementation (``_cpd_hip``, hipcc) sharing one numerics header, so CPU and
GPU results are bit-identical.  Dispatch is by tensor device.

On a machine with a GPU, a missing/failed HIP extension raises immediately —
GPU work silently falling back to an eager emulation is exactly the failure
mode this layer is designed to prevent.
"""
import importlib
import os
import sys

import torch

_HERE = os.path.dirname(os.path.abspath(__file__))

_cpu = None
_hip = None
_hip_err = None


def _import_inplace(name):
    if _HERE not in sys.path:
        sys.path.insert(0, _HERE)
    return importlib.import_module(name)


def _load():
    global _cpu, _hip, _hip_err
    if _cpu is None:
        try:
            _cpu = _import_inplace("_cpd_cpu")
        except ImportError:
            from . import build

            build.build_cpu()
            _cpu = _import_inplace("_cpd_cpu")
    if _hip is None and _hip_err is None:
        try:
            _hip = _import_inplace("_cpd_hip")
        except ImportError as e:
            _hip_err = e
    return _cpu


def cpu_ext():
    _load()
    return _cpu


def hip_ext():
    """The HIP extension module; raises loudly if it is not available."""
    _load()
    if _hip is None:
        raise RuntimeError(
            "cpd_amd HIP extension (_cpd_hip.so) is not available on a GPU "
            "machine — build it with `python -m cpd_amd.ops.build`.  Refusing "
            f"to fall back to eager emulation. Import error: {_hip_err}"
        )
    return _hip


def ext_for(t: torch.Tensor):
    return hip_ext() if t.is_cuda else cpu_ext()


# ---------------------------------------------------------------------------
# op wrappers (contiguity/dtype handled here; extensions assume f32 contig)
# ---------------------------------------------------------------------------

def _f32c(x):
    return x.contiguous().float() if x.dtype != torch.float32 else x.contiguous()


def quantize(x, man_bits, exp_bits):
    x = _f32c(x)
    return ext_for(x).quantize(x, man_bits, exp_bits)


def quantize_(x, man_bits, exp_bits):
    return ext_for(x).quantize_(x, man_bits, exp_bits)


def qadd_(acc, inc, man_bits, exp_bits):
    return ext_for(acc).qadd_(acc, inc.contiguous(), man_bits, exp_bits)


def kahan_qadd_(acc, comp, inc, man_bits, exp_bits):
    return ext_for(acc).kahan_qadd_(acc, comp, inc.contiguous(), man_bits, exp_bits)


def seg_max_exp(flat, offsets, world_size, aligned=False):
    # `aligned=True` (every segment boundary 256-element aligned, n%256==0,
    # guaranteed by GradBucket) selects the wave-uniform vectorized GPU path.
    if flat.is_cuda:
        return hip_ext().seg_max_exp(flat, offsets, world_size, aligned)
    return cpu_ext().seg_max_exp(flat, offsets, world_size)


def scale_quantize_(flat, offsets, shifts, man_bits, exp_bits, aligned=False):
    if flat.is_cuda:
        return hip_ext().scale_quantize_(flat, offsets, shifts, man_bits,
                                         exp_bits, aligned)
    return cpu_ext().scale_quantize_(flat, offsets, shifts, man_bits, exp_bits)


def seg_scale_(flat, offsets, shifts, sign, aligned=False):
    if flat.is_cuda:
        return hip_ext().seg_scale_(flat, offsets, shifts, sign, aligned)
    return cpu_ext().seg_scale_(flat, offsets, shifts, sign)


def quant_gemm_raw(a, b, man_bits, exp_bits):
    return ext_for(a).quant_gemm(a.contiguous(), b.contiguous(), man_bits, exp_bits)


def ceil_log2(x):
    return ext_for(x).ceil_log2(_f32c(x))
