"""HIP extension loading + layout helpers.

The compute path on GPU is the in-tree extension ``_c2v_hip`` (built from
ops/csrc by ``setup.py build_ext --inplace`` or ``__graft_entry__.build()``).
On a CUDA/ROCm device the extension is REQUIRED — ops fail loudly rather
than falling back to eager PyTorch, so a silent-fallback run cannot
masquerade as the native path.

Layout conventions (all HIP kernels assume these):
- wavefront-friendly padding: embed/encode dims are padded to PAD=32-element
  multiples (bf16: 64-byte rows), so every row is 16-B vector aligned,
- the combiner input is packed [B*C, KP] with segment offsets
  [0, TS, TS+PS] for (start-, path-, end-embeddings); KP = 2*TS + PS,
- pad regions are identically zero in parameters, activations and grads
  (asserted by the padding checks in tests/test_model_gpu.py).
"""

from __future__ import annotations

import importlib

PAD = 32  # element granularity of the encode dim (MFMA N tiles)
SEG_PAD = 8  # element granularity of embedding rows / concat segments (16 B)


def round_up(x: int, m: int = PAD) -> int:
    return (x + m - 1) // m * m


def seg_round(x: int) -> int:
    """Embedding-table row stride / concat-segment width: 16-B granules
    (vectorized gathers) without the 32-element MFMA padding waste —
    dt=100 stores as 104, not 128.  The combiner K dim pads the SUM of
    segments up to 32 (KP = round_up(2*TS+PS, 32))."""
    return (x + SEG_PAD - 1) // SEG_PAD * SEG_PAD


_ext = None
_ext_err: Exception | None = None


def _try_load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return
    try:
        import torch  # noqa: F401  (the extension links libtorch/libc10)

        _ext = importlib.import_module("code2vec_amd.ops._c2v_hip")
    except Exception as e:  # noqa: BLE001
        _ext_err = e


def extension_available() -> bool:
    _try_load()
    return _ext is not None


def ext():
    """Return the HIP extension module, or raise loudly.

    Called by every GPU op; never silently substitutes eager PyTorch.
    """
    _try_load()
    if _ext is None:
        raise RuntimeError(
            "code2vec_amd HIP extension (_c2v_hip) is not built/loadable: "
            f"{_ext_err!r}. Build it with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950) or `python -c 'import __graft_entry__; "
            "__graft_entry__.build()'`. The GPU path refuses to fall back to "
            "eager PyTorch."
        )
    return _ext
