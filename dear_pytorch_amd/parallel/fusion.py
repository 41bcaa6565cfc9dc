"""Tensor-fusion layer: gradient bucket planning + HBM-resident fused buffers.

Reference capability: dear/tensorfusion.py (TensorGroup push/pull) +
dear/dopt_rsag.py:90-190 (threshold/nearby-layer grouping, padded pad/shard
buffers).  MI355X-first redesign:

* **grad-as-bucket-view**: instead of per-parameter ``pad_buffer[s:e].copy_``
  during backward (dopt_rsag.py:254-268) and per-parameter pulls in forward
  (dopt_rsag.py:289-304), each parameter's ``.grad`` IS a view into the fused
  bucket.  Autograd accumulates straight into the padded, %P-aligned HBM
  buffer — zero pack/unpack traffic (288 GB HBM3E makes persistent fused
  buffers for every gradient cheap even for BERT-Large).
* buckets are fp32 (or the param dtype), padded so the reduce-scatter shard
  boundary is 256-B aligned per rank — xGMI/RCCL-friendly.
* momentum / optimizer state for a group lives in one contiguous slab laid
  out at the same offsets as the bucket, so the fused update kernel walks
  flat arrays (ops/fused.py, csrc/kernels.hip).

The plan is static after init; per-group chunk descriptors for the fused
kernels are built once and kept on-device.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

__all__ = ["ParamSlot", "BucketGroup", "build_groups", "named_trainable_params"]

ALIGN_ELEMS = 64  # 256 B / 4 B fp32: per-param offset alignment inside a bucket


def _align(n: int, a: int = ALIGN_ELEMS) -> int:
    return (n + a - 1) // a * a


def _is_dense(p: torch.Tensor) -> bool:
    """True when p's strides are a permutation of a contiguous layout
    (e.g. channels_last) — storage covers exactly numel elements."""
    if p.is_contiguous():
        return True
    seen = sorted((s, d) for s, d in zip(p.stride(), p.shape) if d > 1)
    expect = 1
    for s, d in seen:
        if s != expect:
            return False
        expect *= d
    return True


def grad_view(bucket_slice: torch.Tensor, p: torch.Tensor) -> torch.Tensor:
    """View of the flat bucket slice with p's exact layout, so autograd
    accumulates IN-PLACE into the bucket for contiguous AND channels_last
    params (the fused kernels pair bucket[i] with p's storage element i,
    which this layout match guarantees)."""
    if p.is_contiguous():
        return bucket_slice.view(p.shape)
    assert _is_dense(p), "non-dense parameter layout unsupported"
    return bucket_slice.as_strided(p.shape, p.stride())


@dataclass
class ParamSlot:
    name: str
    param: torch.nn.Parameter
    offset: int          # element offset inside the bucket
    numel: int


@dataclass
class BucketGroup:
    index: int
    modules: List[nn.Module]
    slots: List[ParamSlot]
    numel: int = 0                 # used elements incl. inter-param padding
    padded: int = 0                # bucket length (multiple of P * ALIGN)
    bucket: Optional[torch.Tensor] = None
    shard: Optional[torch.Tensor] = None
    extra: dict = field(default_factory=dict)   # optimizer-state slabs etc.

    @property
    def nbytes(self) -> int:
        return self.padded * self.bucket.element_size() if self.bucket is not None else 0

    def allocate(self, world_size: int, device, dtype=torch.float32,
                 comm_dtype=None, attach_grads: bool = True):
        """Allocate the fused bucket + shard and point every param.grad at its
        slice.  With comm_dtype (bf16/fp16) a reduced-precision wire buffer is
        allocated alongside: collectives move half the xGMI bytes while
        accumulation stays fp32 (cast via one dtype-converting copy each way,
        ~HBM-speed, negligible vs the comm saved).

        attach_grads=False (packed-grad mode, dear.py): grads stay None so
        autograd ASSIGNS fresh tensors and one pack_add kernel per group folds
        them into the bucket — instead of one add kernel per parameter."""
        self.padded = _align(max(self.numel, 1), ALIGN_ELEMS * world_size)
        self.bucket = torch.zeros(self.padded, device=device, dtype=dtype)
        shard_n = self.padded // world_size
        if comm_dtype is not None and comm_dtype != dtype:
            self.extra["comm_buf"] = torch.zeros(self.padded, device=device,
                                                 dtype=comm_dtype)
            self.shard = torch.empty(shard_n, device=device, dtype=comm_dtype)
        else:
            self.shard = torch.empty(shard_n, device=device, dtype=dtype)
        for s in self.slots:
            s.param.grad = grad_view(
                self.bucket[s.offset: s.offset + s.numel], s.param) \
                if attach_grads else None

    def reattach_grads(self):
        """Re-point param.grad at bucket views (after anything detached them)."""
        for s in self.slots:
            s.param.grad = grad_view(
                self.bucket[s.offset: s.offset + s.numel], s.param)

    def free(self):
        for s in self.slots:
            s.param.grad = None
        self.bucket = None
        self.shard = None
        self.extra.clear()


def named_trainable_params(model: nn.Module) -> Dict[str, torch.nn.Parameter]:
    return {n: p for n, p in model.named_parameters() if p.requires_grad}


def _module_param_order(model: nn.Module) -> List[Tuple[nn.Module, List[Tuple[str, nn.Parameter]]]]:
    """(module, direct trainable params) in forward (registration) order, with
    shared parameters deduped onto their first owner (reference behavior:
    dopt_rsag.py:206-215 dedupes e.g. BERT tied embeddings)."""
    seen = set()
    out = []
    pname = {p: n for n, p in model.named_parameters(remove_duplicate=False)}
    for m in model.modules():
        direct = []
        for _, p in m.named_parameters(recurse=False):
            if not p.requires_grad or id(p) in seen:
                continue
            seen.add(id(p))
            direct.append((pname.get(p, f"param_{len(seen)}"), p))
        if direct:
            out.append((m, direct))
    return out


def build_groups(model: nn.Module,
                 threshold_bytes: Optional[int] = 25 * 1024 * 1024,
                 num_groups: int = 0,
                 fusion_flags: Optional[List[bool]] = None,
                 nearby_layers: Optional[int] = None) -> List[BucketGroup]:
    """Partition the model's modules (forward order) into bucket groups.

    threshold_bytes: close a group once it holds >= threshold bytes of grads
        (reference THRESHOLD=25 MB, dopt_rsag.py:38).  None => one module per
        group (the no-tensor-fusion ablation).
    num_groups: if > 0, override threshold and split into ~equal-size groups
        (reference num_groups arg, dopt_rsag.py:105-117).
    nearby_layers: with threshold_bytes=None, merge fixed runs of N
        consecutive modules per group (reference NUM_NEARBY_LAYERS,
        dopt_rsag.py:39; N=1 == per-module, the no-fusion ablation).
    fusion_flags: explicit per-module "start new group" booleans (wait-time
        adaptive regrouping, dopt_rsag_wt.py) — length = #modules-with-params.
    """
    mods = _module_param_order(model)
    elem = 4  # plan in fp32 bytes
    if num_groups and num_groups > 0:
        total = sum(p.numel() for _, ps in mods for _, p in ps) * elem
        threshold_bytes = max(total // num_groups, 1)

    groups: List[BucketGroup] = []
    cur: Optional[BucketGroup] = None
    cur_bytes = 0
    cur_mods = 0
    for i, (m, ps) in enumerate(mods):
        start_new = cur is None
        if fusion_flags is not None:
            start_new = start_new or fusion_flags[i]
        elif threshold_bytes is None:
            if nearby_layers and nearby_layers > 1:
                start_new = start_new or cur_mods >= nearby_layers
            else:
                start_new = True
        elif cur_bytes >= threshold_bytes:
            start_new = True
        if start_new:
            cur = BucketGroup(index=len(groups), modules=[], slots=[])
            groups.append(cur)
            cur_bytes = 0
            cur_mods = 0
        cur.modules.append(m)
        cur_mods += 1
        for name, p in ps:
            off = _align(cur.numel)
            cur.slots.append(ParamSlot(name, p, off, p.numel()))
            cur.numel = off + p.numel()
            cur_bytes += p.numel() * elem
    return groups
