"""Synthetic Criteo-shaped sparse data generator.

BASELINE.json benchmarks run on "synthetic Criteo-1TB-shaped sparse" data:
39 fields per row (13 numeric-style + 26 categorical-style), one feature per
field, hashed feature ids in a large global id space with a skewed (power-law)
per-field popularity distribution, random-init weights. Labels are drawn from
a planted sparse-linear teacher so that training has a learnable signal and
test AUC is meaningful (> 0.5 and rising), matching how the reference's
bundled datasets behave under its FM trainer.

Generation happens on-device (GPU when available) so the data pipeline never
bottlenecks the benchmark; `data="synthetic"` is reported by bench.py.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

CRITEO_NUM_FIELDS = 39
# Per-field cardinality profile loosely shaped like Criteo: numeric fields are
# bucketized to small cardinalities; categorical fields range from tiny to
# multi-million. Scaled so the default total id space is ~16.7M (2^24).
_CARD_PROFILE = (
    [64] * 13  # bucketized numeric
    + [int(c) for c in [
        1 << 22, 1 << 21, 1 << 20, 1 << 20, 1 << 19, 1 << 18, 1 << 17,
        1 << 16, 1 << 16, 1 << 15, 1 << 14, 1 << 13, 1 << 13, 1 << 12,
        1 << 12, 1 << 11, 1 << 10, 1 << 10, 256, 256, 128, 128, 64, 32, 16, 8,
    ]]
)


@dataclass
class SyntheticCriteo:
    """Stateful generator; yields CSR-shaped batches with fixed nnz/row."""

    num_features: int = 1 << 24
    num_fields: int = CRITEO_NUM_FIELDS
    seed: int = 1234
    device: str = "cpu"
    teacher_dim: int = 1 << 16  # hashed teacher table size

    def __post_init__(self):
        dev = torch.device(self.device)
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        cards = torch.tensor(_CARD_PROFILE[: self.num_fields], dtype=torch.float64)
        # scale cardinalities to fill num_features
        scale = self.num_features / float(cards.sum())
        cards = (cards * scale).clamp(min=2).to(torch.int64)
        excess = int(cards.sum()) - self.num_features
        if excess > 0:  # clamping small fields can overshoot; shrink largest
            big = int(cards.argmax())
            cards[big] -= excess
            assert cards[big] >= 2, "num_features too small for field profile"
        offsets = torch.zeros(self.num_fields + 1, dtype=torch.int64)
        offsets[1:] = torch.cumsum(cards, 0)
        self.field_cards = cards.to(dev)
        self.field_offsets = offsets.to(dev)
        self.total_ids = int(offsets[-1].item())
        assert self.total_ids <= self.num_features
        # planted teacher: per-hashed-id logit contribution
        self.teacher = (torch.rand(self.teacher_dim, generator=g) - 0.5) * 2.0
        self.teacher = self.teacher.to(dev)
        self._gen = torch.Generator(device=dev if dev.type == "cuda" else "cpu")
        self._gen.manual_seed(self.seed + 1)
        self._dev = dev

    def batch(self, batch_size: int):
        """Returns (row_ptr i32 [B+1], fields i32 [nnz], fids i32 [nnz],
        vals f32 [nnz], labels f32 [B]) on self.device; nnz = B*num_fields."""
        B, F = batch_size, self.num_fields
        dev = self._dev
        u = torch.rand(B, F, generator=self._gen, device=self._gen.device).to(dev)
        # power-law skew within each field: id = floor(card * u^2) concentrates
        # mass on low ids (hot features), like real CTR traffic.
        local = (self.field_cards.view(1, F).to(torch.float32) * u * u).to(torch.int64)
        local = torch.minimum(local, (self.field_cards - 1).view(1, F))
        fids64 = self.field_offsets[:F].view(1, F) + local
        # teacher labels on hashed ids (golden-ratio multiplicative hash)
        h = ((fids64 * 2654435761) >> 8) & (self.teacher_dim - 1)
        logits = self.teacher[h].sum(dim=1) * (3.0 / F**0.5)
        labels = (
            torch.rand(B, generator=self._gen, device=self._gen.device).to(dev)
            < torch.sigmoid(logits)
        ).to(torch.float32)
        row_ptr = torch.arange(0, (B + 1) * F, F, dtype=torch.int32, device=dev)
        fields = (
            torch.arange(F, dtype=torch.int32, device=dev).view(1, F).expand(B, F).reshape(-1)
        )
        vals = torch.ones(B * F, dtype=torch.float32, device=dev)
        return row_ptr, fields.contiguous(), fids64.to(torch.int32).reshape(-1), vals, labels


def synthetic_criteo_batch(
    batch_size: int,
    num_features: int = 1 << 24,
    num_fields: int = CRITEO_NUM_FIELDS,
    seed: int = 1234,
    device: str = "cpu",
):
    return SyntheticCriteo(num_features, num_fields, seed, device).batch(batch_size)
