"""Ring (collective) data parallelism over RCCL/xGMI.

Replaces the reference's hand-rolled Worker_RingReduce
(/root/reference/LightCTR/distribut/ring_collect.h: segmented
reduce-scatter + all-gather over a ZeroMQ worker ring, initializer
broadcast, 1/N averaging) with RCCL collectives: the fused-gradient-slab
design of the reference's BufferFusion (buffer_fusion.h:67-80) maps to
bucketed flat all-reduce; syncInitializer maps to broadcast.

Works with any model exposing dense parameter/grad tensor pairs (e.g. the
MLP layers of NFM/Wide&Deep, CNN, LSTM); gradient averaging by 1/N matches
ring_collect.h:60-68.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def _flatten_into(bucket: torch.Tensor, tensors) -> None:
    off = 0
    for t in tensors:
        n = t.numel()
        bucket[off:off + n].copy_(t.reshape(-1))
        off += n


def _unflatten_from(bucket: torch.Tensor, tensors) -> None:
    off = 0
    for t in tensors:
        n = t.numel()
        t.reshape(-1).copy_(bucket[off:off + n])
        off += n


def broadcast_params(tensors, src: int = 0, group=None) -> None:
    """syncInitializer equivalent (ring_collect.h:74-79): rank src's params
    become everyone's initial params (fused into one broadcast)."""
    tensors = list(tensors)
    if not tensors:
        return
    total = sum(t.numel() for t in tensors)
    bucket = torch.empty(total, dtype=tensors[0].dtype,
                         device=tensors[0].device)
    _flatten_into(bucket, tensors)
    dist.broadcast(bucket, src=src, group=group)
    _unflatten_from(bucket, tensors)


def allreduce_gradients(grads, group=None, bucket_mb: float = 25.0) -> None:
    """syncGradient equivalent (ring_collect.h:48-72): sum-all-reduce the
    fused gradient slab and average by 1/N, bucketed so large models overlap
    transfers. In-place on the given gradient tensors."""
    grads = list(grads)
    if not grads:
        return
    world = dist.get_world_size(group)
    bucket_elems = int(bucket_mb * 1e6 / 4)
    i = 0
    while i < len(grads):
        j, total = i, 0
        while j < len(grads) and (total == 0
                                  or total + grads[j].numel() <= bucket_elems):
            total += grads[j].numel()
            j += 1
        chunk = grads[i:j]
        bucket = torch.empty(total, dtype=chunk[0].dtype,
                             device=chunk[0].device)
        _flatten_into(bucket, chunk)
        dist.all_reduce(bucket, op=dist.ReduceOp.SUM, group=group)
        bucket.div_(world)
        _unflatten_from(bucket, chunk)
        i = j


class RingDataParallel:
    """Wraps a model object exposing `dense_params()` -> [(param, grad)]
    (and optionally `sparse_slabs()`); synchronizes init by broadcast and
    gradients by bucketed all-reduce each step.

    Usage with MLP-bearing models:
        rdp = RingDataParallel(model.mlp_param_grads())  # list provider
        rdp.sync_init()
        ... after model backward, before apply_grads():
        rdp.sync_gradients()
    """

    def __init__(self, param_grad_provider, group=None):
        self._provider = param_grad_provider
        self.group = group

    def sync_init(self) -> None:
        broadcast_params([p for p, _ in self._provider()], src=0,
                         group=self.group)

    def sync_gradients(self) -> None:
        allreduce_gradients([g for _, g in self._provider()],
                            group=self.group)


def mlp_param_grads(mlp):
    """Provider for models.mlp.MLP: weight/bias params with their grads
    (valid after backward)."""

    def provider():
        out = []
        for la in mlp.layers:
            # grads exist only after a backward pass; sync_init (params
            # only) may run before the first step
            out.append((la.W, getattr(la, "_dW", None)))
            out.append((la.b, getattr(la, "_db", None)))
        return out

    return provider


def cnn_param_grads(cnn):
    """Provider for models.cnn.CNNModel — the reference's ring-mode CNN
    (train_cnn_algo.h:64-70,91-97: registerInitializer/registerGradient
    walking the layer chain) expressed as a (param, grad) walk over the
    conv stack + head MLP."""

    def provider():
        out = []
        for la in cnn.layers:
            fc = getattr(la, "fc", None)
            if fc is not None:
                out.append((fc.W, getattr(fc, "_dW", None)))
                out.append((fc.b, getattr(fc, "_db", None)))
        for dl in cnn.head.layers:
            out.append((dl.W, getattr(dl, "_dW", None)))
            out.append((dl.b, getattr(dl, "_db", None)))
        return out

    return provider


def mlp_params(mlp):
    out = []
    for la in mlp.layers:
        out.append(la.W)
        out.append(la.b)
    return out
