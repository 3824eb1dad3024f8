"""Synthetic RGB(+depth)+mask generator.

BASELINE.json:5 replaces the reference's DUTS-style image/mask folder
dataloader (SURVEY §2.1 utils/dataloader.py + joint_transforms.py) with a
seeded synthetic 352x352 RGB + blob-mask generator so random-init training
runs end-to-end with no datasets on disk.

Design goals:
  * deterministic per (seed, index) — the validation split is bit-identical
    across ranks and runs, which is what makes "MAE parity" a meaningful gate;
  * structured, not white noise: each sample is a smooth colored background
    plus K soft blobs; the mask is the union of blob supports.  A CNN can
    actually fit this (loss decreases), unlike pure-noise targets;
  * cheap: generated on CPU in the dataloader workers, torch-only math.

The "joint transforms" of the reference (resize / h-flip / rotate) are folded
in as deterministic per-index augmentation choices of the generator itself.
"""

from __future__ import annotations

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler, Sampler


def _blob_image(g: torch.Generator, size: int, with_depth: bool):
    """One synthetic sample: (rgb[3,S,S], mask[1,S,S]) float32 in [0,1]."""
    S = size
    ys = torch.linspace(-1.0, 1.0, S)
    xs = torch.linspace(-1.0, 1.0, S)
    yy, xx = torch.meshgrid(ys, xs, indexing="ij")

    # smooth background: random low-frequency color gradients
    bg = torch.empty(3, S, S)
    for c in range(3):
        a, b, ph = torch.rand(3, generator=g) * 2 - 1
        f = (torch.rand(1, generator=g).item() * 2.0 + 0.5)
        bg[c] = 0.5 + 0.25 * (a * torch.sin(f * (xx + ph)) + b * torch.cos(f * (yy - ph)))

    nblob = int(torch.randint(1, 4, (1,), generator=g).item())
    mask = torch.zeros(S, S)
    fg = torch.zeros(3, S, S)
    for _ in range(nblob):
        cx, cy = (torch.rand(2, generator=g) * 1.4 - 0.7).tolist()
        rx = 0.12 + torch.rand(1, generator=g).item() * 0.35
        ry = 0.12 + torch.rand(1, generator=g).item() * 0.35
        th = torch.rand(1, generator=g).item() * 3.14159
        ct, st = torch.cos(torch.tensor(th)), torch.sin(torch.tensor(th))
        u = (xx - cx) * ct + (yy - cy) * st
        v = -(xx - cx) * st + (yy - cy) * ct
        d2 = (u / rx) ** 2 + (v / ry) ** 2
        m = (d2 < 1.0).float()
        mask = torch.maximum(mask, m)
        col = torch.rand(3, 1, 1, generator=g)
        soft = torch.clamp(1.0 - d2, 0, 1).sqrt()
        fg += col * (soft * m).unsqueeze(0)

    rgb = torch.clamp(bg * (1 - mask) + fg + 0.05 * torch.randn(3, S, S, generator=g), 0, 1)
    out = [rgb, mask.unsqueeze(0)]
    if with_depth:
        # depth: blobs nearer (brighter) than background, plus a vertical ramp
        depth = torch.clamp(0.3 + 0.4 * (yy + 1) / 2 * (1 - mask) + 0.8 * mask
                            + 0.02 * torch.randn(S, S, generator=g), 0, 1)
        out.insert(1, depth.unsqueeze(0))
    return out


class SyntheticSOD(Dataset):
    """Deterministic synthetic SOD dataset.

    ``split`` selects a disjoint seed space; ``val`` is the fixed-seed
    validation split used for the MAE-parity gate.
    Returns (image, mask) or (image, depth, mask) float32 CHW tensors plus a
    name string (parity with the reference test loader's (img, mask, name)).
    """

    def __init__(self, split: str, size: int, length: int, seed: int = 20260913,
                 with_depth: bool = False):
        assert split in ("train", "val")
        self.split = split
        self.size = size
        self.length = length
        self.seed = seed
        self.with_depth = with_depth
        self._epoch = 0

    def set_epoch(self, epoch: int) -> None:
        # train samples vary per epoch (fresh augmentation space); val is fixed
        self._epoch = epoch if self.split == "train" else 0

    def __len__(self) -> int:
        return self.length

    def __getitem__(self, idx: int):
        base = 0x5EED if self.split == "train" else 0xA11
        g = torch.Generator()
        g.manual_seed((self.seed * 1_000_003 + base) ^ (self._epoch * 7_919 + idx * 2_654_435_761) & 0x7FFFFFFF)
        parts = _blob_image(g, self.size, self.with_depth)
        name = f"{self.split}_{idx:06d}"
        return (*parts, name)


class EvalShardSampler(Sampler):
    """Uneven-shard eval sampler: rank takes ``indices[rank::world]`` with
    NO padding.  A padded DistributedSampler (drop_last=False) duplicates up
    to world-1 samples, which double-counts images in the eval meters; here
    each val image is visited by exactly one rank exactly once."""

    def __init__(self, length: int, rank: int, world_size: int):
        self._idx = range(rank, length, world_size)

    def __iter__(self):
        return iter(self._idx)

    def __len__(self) -> int:
        return len(self._idx)

    def set_epoch(self, epoch: int) -> None:  # API parity; val order is fixed
        pass


def create_loader(cfg, split: str, distributed: bool, rank: int = 0,
                  world_size: int = 1) -> tuple[DataLoader, "DistributedSampler | None"]:
    """Reference-parity loader factory (SURVEY §2.1 create_loader)."""
    with_depth = cfg.model.startswith("hdfnet")
    length = cfg.train_samples if split == "train" else cfg.val_samples
    ds = SyntheticSOD(split, cfg.size, length, seed=cfg.data_seed, with_depth=with_depth)
    shuffle = getattr(cfg, "shuffle", True)
    sampler = None
    if distributed:
        if split == "train":
            # shuffle=False shards indices[rank::world]: the union of rank
            # batches at step k is exactly the 1-rank batch k (parity runs)
            sampler = DistributedSampler(ds, num_replicas=world_size,
                                         rank=rank, shuffle=shuffle,
                                         drop_last=True)
        else:
            sampler = EvalShardSampler(length, rank, world_size)
    loader = DataLoader(
        ds,
        batch_size=cfg.batch_size,
        shuffle=(sampler is None and split == "train" and shuffle),
        sampler=sampler,
        num_workers=cfg.num_workers,
        pin_memory=torch.cuda.is_available(),
        drop_last=(split == "train"),
        # NOT persistent: live workers hold a pickled dataset copy, so the
        # per-epoch set_epoch() (fresh augmentation space) would never
        # reach them and every epoch would re-draw epoch-0 samples.
        # Respawning workers each epoch pickles the post-set_epoch state.
        persistent_workers=False,
    )
    return loader, sampler
