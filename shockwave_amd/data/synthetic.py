"""Synthetic in-memory datasets shaped like the reference workloads' data.

There is no network on the build/GPU boxes, so every workload trains on
random tensors with the real datasets' shapes and cardinalities
(dataset sizes: core/data/job_profiles.json; BASELINE requires
``data: synthetic``).
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset


class SyntheticImages(Dataset):
    """CIFAR-10 (32x32) or ImageNet (224x224) shaped images."""

    def __init__(self, num_samples, image_size=32, num_classes=10, seed=0):
        self.num_samples = num_samples
        self.image_size = image_size
        self.num_classes = num_classes
        g = torch.Generator().manual_seed(seed)
        # small resident pool re-indexed modulo, so memory (and per-job
        # startup time) stays bounded: 2048 images at 32x32, 256 at 224x224
        pool = min(num_samples, 2048 if image_size <= 64 else 256)
        self.images = torch.randn(pool, 3, image_size, image_size, generator=g)
        self.labels = torch.randint(
            0, num_classes, (pool,), generator=g
        )

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        i = idx % self.images.size(0)
        return self.images[i], self.labels[i]


class SyntheticTranslation(Dataset):
    """Multi30k-shaped (short sentence pairs)."""

    def __init__(self, num_samples=10000, src_vocab=9521, tgt_vocab=17851,
                 src_len=24, tgt_len=26, seed=0):
        g = torch.Generator().manual_seed(seed)
        pool = min(num_samples, 4096)
        self.src = torch.randint(4, src_vocab, (pool, src_len), generator=g)
        self.tgt = torch.randint(4, tgt_vocab, (pool, tgt_len), generator=g)
        self.num_samples = num_samples

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        i = idx % self.src.size(0)
        return self.src[i], self.tgt[i]


class SyntheticCorpus:
    """Wikitext-2-shaped token stream for the LSTM LM (batchified access,
    like the stock word_language_model pipeline)."""

    def __init__(self, num_tokens=2088628, vocab=33278, bptt=35, seed=0):
        g = torch.Generator().manual_seed(seed)
        pool = min(num_tokens, 1 << 20)
        self.tokens = torch.randint(0, vocab, (pool,), generator=g)
        self.num_tokens = num_tokens
        self.bptt = bptt
        self.vocab = vocab

    def batchify(self, batch_size):
        n = self.tokens.size(0) // batch_size
        return self.tokens[: n * batch_size].view(batch_size, -1).t().contiguous()


class SyntheticInteractions(Dataset):
    """ML-20M-shaped sparse user-item interaction vectors (dense here)."""

    def __init__(self, num_users=117907, num_items=20108, density=0.005,
                 seed=0):
        self.num_users = num_users
        self.num_items = num_items
        g = torch.Generator().manual_seed(seed)
        pool = min(num_users, 512)
        self.rows = (
            torch.rand(pool, num_items, generator=g) < density
        ).float()

    def __len__(self):
        return self.num_users

    def __getitem__(self, idx):
        return self.rows[idx % self.rows.size(0)]


class SyntheticUnpairedImages(Dataset):
    """monet2photo-shaped image pairs for CycleGAN."""

    def __init__(self, num_samples=6287, image_size=128, seed=0):
        g = torch.Generator().manual_seed(seed)
        pool = min(num_samples, 256)
        self.a = torch.randn(pool, 3, image_size, image_size, generator=g)
        self.b = torch.randn(pool, 3, image_size, image_size, generator=g)
        self.num_samples = num_samples

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        i = idx % self.a.size(0)
        return {"A": self.a[i], "B": self.b[i]}
