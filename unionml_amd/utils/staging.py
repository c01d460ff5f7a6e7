"""Pinned-buffer H2D staging for parsed splits (SURVEY.md §2c row
'pinned staging loader').

Feature matrices come out of the dataset pipeline as DataFrames or
ndarrays; this module turns them into device tensors by way of a
*reused* pinned host buffer and an async H2D copy on a dedicated copy
stream, so staging overlaps compute. With 288 GB of HBM3E per MI355X
the whole dataset is staged resident — tensors are uploaded once and
re-read from HBM, never re-staged per epoch.
"""

from typing import Any, List, Optional

import numpy as np


class PinnedStager:
    """Reusable pinned host buffer + copy stream for one device."""

    def __init__(self, device=None):
        import torch

        self.torch = torch
        self.device = torch.device(device) if device is not None else torch.device(
            "cuda" if torch.cuda.is_available() else "cpu"
        )
        self._pinned: dict = {}  # dtype -> host buffer, grown geometrically
        self._inflight: dict = {}  # dtype -> event of the last async copy FROM the buffer
        self.copy_stream = (
            torch.cuda.Stream(device=self.device) if self.device.type == "cuda" else None
        )

    def _pinned_buffer(self, numel: int, dtype):
        # the previous async H2D reading this buffer must have completed
        # before the host overwrites it (wait_stream orders GPU streams,
        # not host writes — without this, back-to-back stagings race)
        ev = self._inflight.pop(dtype, None)
        if ev is not None:
            ev.synchronize()
        buf = self._pinned.get(dtype)
        if buf is None or buf.numel() < numel:
            cap = max(numel, 2 * buf.numel() if buf is not None else numel)
            buf = self.torch.empty(cap, dtype=dtype, pin_memory=self.device.type == "cuda")
            self._pinned[dtype] = buf
        return buf

    def to_device(self, array: Any, non_blocking: bool = True):
        """array/DataFrame/tensor -> tensor on the target device via the
        pinned buffer (hipMemcpyAsync under the hood on ROCm)."""
        torch = self.torch
        if hasattr(array, "to_numpy"):  # DataFrame / Series
            array = array.to_numpy()
        if isinstance(array, np.ndarray):
            if not array.flags.c_contiguous:
                array = np.ascontiguousarray(array)
            t = torch.from_numpy(array)
        elif torch.is_tensor(array):
            t = array
        else:
            return array  # non-array payloads pass through

        if self.device.type != "cuda":
            return t.clone()

        if t.device.type == "cuda":
            return t.to(self.device)

        flat = t.reshape(-1)
        host = self._pinned_buffer(flat.numel(), t.dtype)[: flat.numel()]
        host.copy_(flat)
        stream = self.copy_stream or torch.cuda.current_stream(self.device)
        with torch.cuda.stream(stream):
            dev = torch.empty(t.shape, dtype=t.dtype, device=self.device)
            dev.reshape(-1).copy_(host, non_blocking=non_blocking)
            ev = torch.cuda.Event()
            ev.record(stream)
            self._inflight[t.dtype] = ev
        if self.copy_stream is not None:
            torch.cuda.current_stream(self.device).wait_stream(self.copy_stream)
            # the tensor was allocated on the copy stream but lives on
            # the compute stream from here: tell the caching allocator,
            # or a later free could recycle the block while compute
            # kernels still read it
            dev.record_stream(torch.cuda.current_stream(self.device))
        return dev


_default_stagers: dict = {}


def get_stager(device=None) -> PinnedStager:
    key = str(device)
    if key not in _default_stagers:
        _default_stagers[key] = PinnedStager(device)
    return _default_stagers[key]


def stage_split_to_device(split: List[Any], device=None, stream=None) -> List[Any]:
    stager = get_stager(device)
    if stream is not None:
        stager.copy_stream = stream
    return [stager.to_device(el) for el in split]
