"""GPU BPE tokenizer: batch packing + kernel dispatch.

Packs a batch of texts into one byte buffer + request offsets, runs the
segmenter + wave-per-segment BPE kernels (csrc/aigw_kernels.hip), and
returns per-request token counts (and optionally ids). The merge table is
uploaded once and stays HBM/L2-resident.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from aigw.ops import load_hip_module
from aigw.ops.bpe_ref import BPERef, build_hash_table, make_merges


class GPUTokenizer:
    def __init__(
        self,
        n_merges: int = 32768,
        seed: int = 1355,
        device: str = "cuda",
        _hip=None,
    ):
        self.merges = make_merges(n_merges, seed)
        self.device = torch.device(device)
        self.hip = _hip if _hip is not None else load_hip_module()
        if self.hip is None:
            raise RuntimeError("aigw_hip extension unavailable")
        keys, ranks = build_hash_table(self.merges)
        self.htab_keys = torch.from_numpy(keys).to(self.device)
        self.htab_ranks = torch.from_numpy(ranks).to(self.device)
        self.vocab_size = 256 + len(self.merges)
        # pinned staging buffers for truly-async H2D (pageable copies block)
        self._pin_bytes = None
        self._pin_offs = None
        self._h2d_ev = None  # guards pinned-buffer reuse vs in-flight DMA

    @staticmethod
    def pack(texts: list[bytes]) -> tuple[np.ndarray, np.ndarray]:
        data = b"".join(texts)
        offs = np.zeros(len(texts), dtype=np.int64)
        off = 0
        for i, t in enumerate(texts):
            offs[i] = off
            off += len(t)
        return np.frombuffer(data, dtype=np.uint8).copy(), offs

    def encode_batch(
        self, texts: list[bytes], return_ids: bool = False
    ) -> tuple[torch.Tensor, Optional[list[list[int]]], dict]:
        """Returns (req_counts int32[n_req] on GPU, ids or None, gpu_state).

        gpu_state carries the on-device tensors (out_ids, req_off) so the
        embedding stage can consume them without a host round-trip.
        """
        arr, offs = self.pack(texts)
        # stage through pinned buffers: pageable H2D copies BLOCK on ROCm
        # (the driver bounces them through an internal staging buffer),
        # which taxed every admission batch on the serving hot path
        bytes_t, off_t = self._stage(arr, offs)
        out_ids, req_counts, seg_start, seg_req = self.hip.bpe_encode(
            bytes_t, off_t, self.htab_keys, self.htab_ranks
        )
        ids = None
        if return_ids:
            ids = []
            flat = out_ids.cpu().numpy()
            bounds = list(offs) + [len(arr)]
            for i in range(len(texts)):
                seg = flat[bounds[i] : bounds[i + 1]]
                ids.append([int(x) for x in seg[seg >= 0]])
        return req_counts, ids, {"out_ids": out_ids, "req_off": off_t, "n_bytes": len(arr)}

    def _stage(self, arr, offs):
        """Copy packed host arrays into reusable pinned buffers and issue
        truly-async H2D copies; returns the device tensors."""
        n = len(arr)
        if self._h2d_ev is not None:
            # the pinned buffers may still be the source of an in-flight
            # copy from the PREVIOUS batch; wait for that DMA (usually
            # already complete) before overwriting them
            self._h2d_ev.synchronize()
        if self._pin_bytes is None or self._pin_bytes.numel() < n:
            self._pin_bytes = torch.empty(max(n, 1 << 20), dtype=torch.uint8,
                                          pin_memory=True)
        if self._pin_offs is None or self._pin_offs.numel() < len(offs):
            self._pin_offs = torch.empty(max(len(offs), 4096), dtype=torch.int64,
                                         pin_memory=True)
        self._pin_bytes.numpy()[:n] = arr
        self._pin_offs.numpy()[: len(offs)] = offs
        bytes_t = self._pin_bytes[:n].to(self.device, non_blocking=True)
        off_t = self._pin_offs[: len(offs)].to(self.device, non_blocking=True)
        self._h2d_ev = torch.cuda.Event()
        self._h2d_ev.record()
        return bytes_t, off_t

    def encode_batch_async(self, texts: list[bytes]):
        """Sync-free batch encode: returns (req_counts_gpu, out_ids_gpu,
        req_off_gpu, event). Await the event (cooperatively, via
        event.query()) before reading the tensors — no host sync happens
        here, which matters because ROCm host syncs busy-spin a core."""
        arr, offs = self.pack(texts)
        bytes_t, off_t = self._stage(arr, offs)
        out_ids, req_counts = self.hip.bpe_count_async(
            bytes_t, off_t, self.htab_keys, self.htab_ranks
        )
        ev = torch.cuda.Event()
        ev.record()
        return req_counts, out_ids, off_t, ev

    def reference(self) -> BPERef:
        return BPERef(self.merges)
