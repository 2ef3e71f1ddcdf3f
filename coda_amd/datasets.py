"""Prediction-tensor datasets.

On-disk format (identical to the reference, coda/datasets.py:4-23): a
`<task>.pt` tensor of shape (H, N, C) holding post-softmax scores - H models,
N points, C classes - optionally with `<task>_labels.pt` holding (N,) integer
ground-truth labels. Stored dtype may be fp16/bf16; compute is always fp32
(the loader up-casts).

Additions over the reference:
  - model-axis sharding for multi-GPU runs (`shard` argument): rank r keeps
    rows h where h % world == r, so every rank's slice is balanced.
  - pinned-host staging for large tensors (`pin=True`): the (H,N,C) tensor is
    loaded to pinned CPU memory and copied to the device on a side stream
    (non_blocking), instead of the reference's single blocking load.
  - synthetic task generator with a planted best model (for tests/benchmarks).
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch


STORAGE_DTYPES = {
    "fp32": torch.float32,
    "bf16": torch.bfloat16,
    "fp8": torch.float8_e4m3fn,
}


class Dataset:
    def __init__(self, filepath: str, device,
                 shard: Optional[Tuple[int, int]] = None,
                 pin: bool = False,
                 storage_dtype: str = "fp32",
                 stream_chunk_mb: int = 0):
        """Load a prediction tensor.

        Args:
            filepath: path to the (H, N, C) .pt file.
            device: target device.
            shard: optional (rank, world_size); keeps models h with
                h % world_size == rank.
            pin: stage through pinned host memory with an async copy.
            storage_dtype: on-device storage format - "fp32" (reference
                parity), "bf16" or "fp8" (e4m3). Compute always up-casts
                to fp32 (the reference's own fp16-on-disk -> .float()
                pattern, coda/datasets.py:14, generalized). Halving /
                quartering storage is what fits large model pools into
                the 288 GB of HBM3E.
            stream_chunk_mb: >0 streams the tensor host->device in chunks
                of this size through pinned buffers on a side stream
                (dtype conversion on device), instead of one blocking copy.
        """
        self.device = torch.device(device)
        preds = torch.load(filepath, map_location="cpu", weights_only=True)
        preds = preds.float()
        self.total_models = preds.shape[0]
        self.shard = shard
        if shard is not None:
            rank, world = shard
            self.model_idxs = torch.arange(rank, self.total_models, world)
            preds = preds[self.model_idxs]
        else:
            self.model_idxs = torch.arange(self.total_models)
        self.storage_dtype = storage_dtype
        dt = STORAGE_DTYPES[storage_dtype]
        if stream_chunk_mb and self.device.type == "cuda":
            self.preds = _stream_to_device(preds, self.device, dt,
                                           stream_chunk_mb)
        else:
            self.preds = _to_device(preds, self.device, pin).to(dt)

        self.labels = None
        label_p = filepath.replace(".pt", "_labels.pt")
        if os.path.exists(label_p):
            labels = torch.load(label_p, map_location="cpu", weights_only=True)
            self.labels = labels.to(self.device)

    @classmethod
    def from_tensors(cls, preds: torch.Tensor, labels: Optional[torch.Tensor],
                     device, shard: Optional[Tuple[int, int]] = None):
        """Build a Dataset from in-memory tensors (tests, synthetic tasks).

        A tensor already in a storage dtype (bf16 / fp8-e4m3) is kept as
        is - up-casting a 1M-point fp8 pool to fp32 would be 512 GB.
        """
        self = cls.__new__(cls)
        self.device = torch.device(device)
        if preds.dtype not in STORAGE_DTYPES.values():
            preds = preds.float()
        self.total_models = preds.shape[0]
        self.shard = shard
        if shard is not None:
            rank, world = shard
            self.model_idxs = torch.arange(rank, self.total_models, world)
            preds = preds[self.model_idxs]
        else:
            self.model_idxs = torch.arange(self.total_models)
        self.preds = preds.to(self.device)
        self.labels = labels.to(self.device) if labels is not None else None
        return self


def _stream_to_device(t: torch.Tensor, device: torch.device,
                      dtype: torch.dtype, chunk_mb: int) -> torch.Tensor:
    """Pinned double-buffered host->HBM streaming with on-device dtype
    conversion (fp32 on the host wire here; a bf16/fp8 source file would
    halve wire bytes too). Chunks along the model axis."""
    H = t.shape[0]
    out = torch.empty(t.shape, dtype=dtype, device=device)
    bytes_per_model = t[0].numel() * t.element_size()
    rows = max(1, (chunk_mb << 20) // max(1, bytes_per_model))
    side = torch.cuda.Stream(device=device)
    bufs = [torch.empty((min(rows, H),) + t.shape[1:],
                        dtype=t.dtype).pin_memory() for _ in range(2)]
    events = [torch.cuda.Event(), torch.cuda.Event()]
    for e in events:
        e.record(side)
    for i, h0 in enumerate(range(0, H, rows)):
        h1 = min(h0 + rows, H)
        buf = bufs[i % 2]
        events[i % 2].synchronize()  # buffer free to reuse
        buf[:h1 - h0].copy_(t[h0:h1])
        with torch.cuda.stream(side):
            staged = buf[:h1 - h0].to(device, non_blocking=True)
            out[h0:h1].copy_(staged.to(dtype))
            events[i % 2].record(side)
    torch.cuda.current_stream(device).wait_stream(side)
    return out


def _to_device(t: torch.Tensor, device: torch.device, pin: bool) -> torch.Tensor:
    if device.type != "cuda" or not pin:
        return t.to(device)
    # Pinned staging + async copy on a side stream; keeps the default stream
    # free while a large pool streams into HBM.
    pinned = t.pin_memory()
    side = torch.cuda.Stream(device=device)
    with torch.cuda.stream(side):
        out = pinned.to(device, non_blocking=True)
    torch.cuda.current_stream(device).wait_stream(side)
    return out


def make_synthetic_task(H: int = 8, N: int = 500, C: int = 10,
                        seed: int = 0, best_acc: float = 0.9,
                        worst_acc: float = 0.5, temperature: float = 3.0):
    """Random (H, N, C) prediction tensor with a planted best model.

    Model h's accuracy interpolates from best_acc (h=0) to worst_acc
    (h=H-1); predictions are softmaxed logits biased toward the model's
    (possibly corrupted) predicted class. Returns (preds, labels) on CPU.
    """
    g = torch.Generator().manual_seed(seed)
    labels = torch.randint(0, C, (N,), generator=g)
    accs = torch.linspace(best_acc, worst_acc, H)
    preds = torch.empty(H, N, C)
    for h in range(H):
        correct = torch.rand(N, generator=g) < accs[h]
        wrong = torch.randint(1, C, (N,), generator=g)
        pred_class = torch.where(correct, labels, (labels + wrong) % C)
        logits = torch.randn(N, C, generator=g)
        logits[torch.arange(N), pred_class] += temperature
        preds[h] = torch.softmax(logits, dim=-1)
    return preds, labels


def write_synthetic_task(data_dir: str, name: str = "synthetic", **kwargs):
    """Materialize a synthetic task as <data_dir>/<name>.pt (+ _labels.pt)."""
    os.makedirs(data_dir, exist_ok=True)
    preds, labels = make_synthetic_task(**kwargs)
    torch.save(preds, os.path.join(data_dir, f"{name}.pt"))
    torch.save(labels, os.path.join(data_dir, f"{name}_labels.pt"))
    return os.path.join(data_dir, f"{name}.pt")
