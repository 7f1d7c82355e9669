"""GPU ops: HIP kernel bindings + CPU reference oracles.

``aigw_hip`` (built from csrc/aigw_kernels.hip for gfx950) is REQUIRED when
a GPU is present — ops fail loudly rather than falling back to eager
PyTorch on a GPU box. On CPU-only hosts the pure-Python oracles in bpe_ref
are used (they are also the numerics reference the GPU tests compare
against).
"""

from __future__ import annotations

import torch


def load_hip_module():
    """Import the in-tree extension; loud failure on GPU hosts."""
    try:
        import aigw_hip  # built in-tree by setup.py build_ext --inplace

        return aigw_hip
    except ImportError as e:
        if torch.cuda.is_available():
            raise RuntimeError(
                "aigw_hip extension not built but a GPU is present. Build it "
                "with: python setup.py build_ext --inplace (PYTORCH_ROCM_ARCH=gfx950). "
                "Refusing to fall back to eager PyTorch on the GPU hot path."
            ) from e
        return None
