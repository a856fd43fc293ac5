"""Dataset ingestion for the benchmark entry points.

Reference parity: the APP=1/2/else dataset blocks that every reference
benchmark repeats (e.g. benchmarks/layer_parallelism/
benchmark_resnet_lp.py:177-231): APP 1 = ImageFolder(datapath) with
shuffle, APP 2 = CIFAR-10 train split without shuffle, anything else =
synthetic fake data of 10*batch samples; all with the transform
``ToTensor() + Normalize((0.5,)*3, (0.5,)*3)`` (pixels -> [-1, 1]).

The reference leans on torchvision for all three; torchvision is not
part of this stack, so the readers are implemented directly — PIL for
image decode, the CIFAR-10 python-pickle batch format via numpy — with
the same semantics. Additions over the reference:

* optional resize to the training resolution (the reference assumes
  the on-disk images already match ``--image-size``; very-high-res
  training wants the dataset reader to guarantee it),
* ``.npy`` / ``.pt`` samples in image folders (common for synthetic
  very-high-resolution corpora where JPEG decode would dominate),
* deterministic shuffling shared by every rank (the reference gets
  cross-rank batch agreement implicitly from ``torch.manual_seed(0)``;
  here it is explicit via a seeded generator).

Every rank reads the full dataset and steps through the same batch
sequence — matching the reference, where model/spatial parallelism
replicates the input pipeline on each rank (the first stage slices
what it needs; see parallel/spatial.py).
"""

from __future__ import annotations

import os
import pickle
from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset

IMG_EXTS = (".jpg", ".jpeg", ".png", ".ppm", ".bmp", ".pgm", ".tif", ".tiff")
TENSOR_EXTS = (".npy", ".pt")


def to_tensor_normalized(arr: np.ndarray) -> torch.Tensor:
    """HWC uint8 -> CHW float32 in [-1, 1] (ToTensor + Normalize(0.5, 0.5))."""
    # copy: PIL hands out read-only arrays and from_numpy aliases
    t = torch.from_numpy(np.array(arr, copy=True)).permute(2, 0, 1).float()
    return t.div_(127.5).sub_(1.0)


def _load_image(path: str, image_size: Optional[int]) -> torch.Tensor:
    ext = os.path.splitext(path)[1].lower()
    if ext == ".npy":
        arr = np.load(path)
        if arr.ndim == 3 and arr.shape[0] in (1, 3):  # CHW -> HWC
            arr = np.moveaxis(arr, 0, -1)
        if arr.dtype == np.uint8:
            x = to_tensor_normalized(arr)
        else:
            x = torch.from_numpy(np.ascontiguousarray(arr)).permute(2, 0, 1).float()
    elif ext == ".pt":
        x = torch.load(path, map_location="cpu", weights_only=True)
        if x.ndim != 3:
            raise ValueError(f"{path}: expected a CHW tensor, got shape {tuple(x.shape)}")
        x = x.float()
    else:
        from PIL import Image

        with Image.open(path) as im:
            im = im.convert("RGB")
            if image_size is not None and im.size != (image_size, image_size):
                im = im.resize((image_size, image_size), Image.BILINEAR)
            arr = np.asarray(im, dtype=np.uint8)
        return to_tensor_normalized(arr)
    if image_size is not None and x.shape[-2:] != (image_size, image_size):
        x = torch.nn.functional.interpolate(
            x.unsqueeze(0), size=(image_size, image_size),
            mode="bilinear", align_corners=False,
        ).squeeze(0)
    return x


class ImageFolderDataset(Dataset):
    """APP=1: one subdirectory per class, any mix of image/tensor files.

    Mirrors torchvision.datasets.ImageFolder as used by the reference
    (benchmark_resnet_lp.py:184-188): classes are the sorted
    subdirectory names, targets their indices.
    """

    def __init__(self, root: str, image_size: Optional[int] = None):
        self.root = root
        self.image_size = image_size
        if not os.path.isdir(root):
            raise FileNotFoundError(f"ImageFolder root not found: {root}")
        self.classes: List[str] = sorted(
            d for d in os.listdir(root) if os.path.isdir(os.path.join(root, d))
        )
        if not self.classes:
            raise ValueError(f"{root}: no class subdirectories")
        self.class_to_idx = {c: i for i, c in enumerate(self.classes)}
        self.samples: List[Tuple[str, int]] = []
        for c in self.classes:
            cdir = os.path.join(root, c)
            for fn in sorted(os.listdir(cdir)):
                if fn.lower().endswith(IMG_EXTS + TENSOR_EXTS):
                    self.samples.append((os.path.join(cdir, fn), self.class_to_idx[c]))
        if not self.samples:
            raise ValueError(f"{root}: no images found (extensions: {IMG_EXTS + TENSOR_EXTS})")

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, i: int):
        path, target = self.samples[i]
        return _load_image(path, self.image_size), target


class CIFAR10Dataset(Dataset):
    """APP=2: the CIFAR-10 python-pickle batch format, read directly.

    The reference uses torchvision.datasets.CIFAR10(download=True)
    (benchmark_resnet_lp.py:198-200); there is no network here, so the
    batches must already exist under ``root`` (or
    ``root/cifar-10-batches-py``). Raises with a clear message if not.
    """

    TRAIN_FILES = [f"data_batch_{i}" for i in range(1, 6)]
    TEST_FILES = ["test_batch"]

    def __init__(self, root: str, train: bool = True,
                 image_size: Optional[int] = None):
        base = root
        sub = os.path.join(root, "cifar-10-batches-py")
        if os.path.isdir(sub):
            base = sub
        names = self.TRAIN_FILES if train else self.TEST_FILES
        paths = [os.path.join(base, n) for n in names]
        found = [p for p in paths if os.path.isfile(p)]
        if not found:
            raise FileNotFoundError(
                f"CIFAR-10 batches not found under {root} — expected "
                f"{names} (no network: place the extracted "
                "cifar-10-batches-py directory there)"
            )
        data, labels = [], []
        for p in found:
            with open(p, "rb") as f:
                d = pickle.load(f, encoding="bytes")
            data.append(np.asarray(d[b"data"], dtype=np.uint8))
            labels.extend(d.get(b"labels", d.get(b"fine_labels", [])))
        self.data = np.concatenate(data).reshape(-1, 3, 32, 32)
        self.targets = list(int(v) for v in labels)
        self.image_size = image_size

    def __len__(self) -> int:
        return len(self.data)

    def __getitem__(self, i: int):
        x = to_tensor_normalized(np.moveaxis(self.data[i], 0, -1))
        if self.image_size is not None and self.image_size != 32:
            x = torch.nn.functional.interpolate(
                x.unsqueeze(0), size=(self.image_size,) * 2,
                mode="bilinear", align_corners=False,
            ).squeeze(0)
        return x, self.targets[i]


class SyntheticDataset(Dataset):
    """APP=3 (default): random images/labels, deterministic per index —
    the role torchvision.datasets.FakeData(random_offset=0) plays in the
    reference (benchmark_resnet_lp.py:210-217)."""

    def __init__(self, size: int, image_shape: Sequence[int], num_classes: int,
                 seed: int = 0):
        self.size = int(size)
        self.image_shape = tuple(image_shape)
        self.num_classes = int(num_classes)
        self.seed = seed

    def __len__(self) -> int:
        return self.size

    def __getitem__(self, i: int):
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + i)
        x = torch.randn(self.image_shape, generator=g)
        y = int(torch.randint(0, self.num_classes, (1,), generator=g))
        return x, y


def make_dataloader(
    app: int,
    datapath: Optional[str],
    batch_size: int,
    image_size: int,
    num_classes: int,
    num_workers: int = 0,
    times: int = 1,
    train: bool = True,
    seed: int = 0,
) -> Tuple[DataLoader, int]:
    """Build the (dataloader, dataset_size) pair with the reference's
    APP wiring: app 1 = shuffled ImageFolder, app 2 = unshuffled
    CIFAR-10, else synthetic 10*batch samples. batch = times*batch_size
    (GEMS runs ``times`` replica pairs per step). drop_last keeps batch
    shapes static — the engines preallocate P2P buffers per micro-batch.
    """
    B = times * batch_size
    if app == 1:
        if not datapath:
            raise ValueError("--app 1 (ImageFolder) requires --datapath")
        ds: Dataset = ImageFolderDataset(datapath, image_size=image_size)
        shuffle = True
    elif app == 2:
        if not datapath:
            raise ValueError("--app 2 (CIFAR-10) requires --datapath")
        ds = CIFAR10Dataset(datapath, train=train, image_size=image_size)
        shuffle = False
    else:
        ds = SyntheticDataset(10 * B, (3, image_size, image_size), num_classes,
                              seed=seed)
        shuffle = False
    gen = torch.Generator().manual_seed(seed) if shuffle else None
    loader = DataLoader(
        ds,
        batch_size=B,
        shuffle=shuffle,
        generator=gen,
        num_workers=num_workers,
        pin_memory=torch.cuda.is_available(),
        drop_last=True,
    )
    return loader, len(ds)
