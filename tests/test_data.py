"""Dataset ingestion: ImageFolder / CIFAR-10 pickle format / synthetic
(reference APP=1/2/else wiring, benchmark_resnet_lp.py:177-231) and the
end-to-end benchmark runner on real data over gloo."""

import os
import pickle

import numpy as np
import pytest
import torch

from dist_util import run_distributed


def _make_image_tree(root, classes=("cat", "dog"), per_class=3, size=16):
    from PIL import Image

    rng = np.random.default_rng(0)
    for c in classes:
        d = os.path.join(root, c)
        os.makedirs(d, exist_ok=True)
        for i in range(per_class):
            arr = rng.integers(0, 256, (size, size, 3), dtype=np.uint8)
            Image.fromarray(arr).save(os.path.join(d, f"{i}.png"))
    return root


def _make_cifar(root, n=8):
    d = os.path.join(root, "cifar-10-batches-py")
    os.makedirs(d, exist_ok=True)
    rng = np.random.default_rng(1)
    data = rng.integers(0, 256, (n, 3072), dtype=np.uint8)
    labels = [int(v) for v in rng.integers(0, 10, n)]
    with open(os.path.join(d, "data_batch_1"), "wb") as f:
        pickle.dump({b"data": data, b"labels": labels}, f)
    return root, data, labels


def test_image_folder(tmp_path):
    from mpi4dl_amd.data import ImageFolderDataset

    root = _make_image_tree(str(tmp_path))
    ds = ImageFolderDataset(root, image_size=16)
    assert ds.classes == ["cat", "dog"]
    assert len(ds) == 6
    x, y = ds[0]
    assert x.shape == (3, 16, 16) and y == 0
    assert x.min() >= -1.0 and x.max() <= 1.0
    # resize path
    ds32 = ImageFolderDataset(root, image_size=32)
    assert ds32[5][0].shape == (3, 32, 32) and ds32[5][1] == 1


def test_image_folder_tensor_files(tmp_path):
    from mpi4dl_amd.data import ImageFolderDataset

    d = tmp_path / "a"
    d.mkdir()
    np.save(d / "x.npy", np.zeros((8, 8, 3), dtype=np.uint8))
    torch.save(torch.ones(3, 8, 8), d / "y.pt")
    ds = ImageFolderDataset(str(tmp_path), image_size=8)
    assert len(ds) == 2
    assert torch.allclose(ds[0][0], torch.full((3, 8, 8), -1.0))  # uint8 0 -> -1
    assert torch.allclose(ds[1][0], torch.ones(3, 8, 8))  # .pt passes through


def test_cifar10_pickle(tmp_path):
    from mpi4dl_amd.data import CIFAR10Dataset

    root, data, labels = _make_cifar(str(tmp_path))
    ds = CIFAR10Dataset(root)
    assert len(ds) == 8
    x, y = ds[3]
    assert x.shape == (3, 32, 32) and y == labels[3]
    # exact value check: pixel (c,h,w) = data row reshaped (3,32,32), in [-1,1]
    ref = torch.from_numpy(data[3].reshape(3, 32, 32)).float() / 127.5 - 1.0
    assert torch.allclose(x, ref)
    up = CIFAR10Dataset(root, image_size=64)
    assert up[0][0].shape == (3, 64, 64)


def test_cifar10_missing(tmp_path):
    from mpi4dl_amd.data import CIFAR10Dataset

    with pytest.raises(FileNotFoundError):
        CIFAR10Dataset(str(tmp_path))


def test_synthetic_deterministic():
    from mpi4dl_amd.data import SyntheticDataset

    ds = SyntheticDataset(10, (3, 8, 8), 5)
    x1, y1 = ds[3]
    x2, y2 = ds[3]
    assert torch.equal(x1, x2) and y1 == y2
    assert not torch.equal(x1, ds[4][0])


def test_make_dataloader_wiring(tmp_path):
    from mpi4dl_amd.data import make_dataloader

    root, _, _ = _make_cifar(str(tmp_path))
    loader, n = make_dataloader(2, root, batch_size=3, image_size=32,
                                num_classes=10)
    assert n == 8
    xb, yb = next(iter(loader))
    assert xb.shape == (3, 3, 32, 32) and yb.shape == (3,)
    assert len(loader) == 2  # drop_last: 8 // 3

    loader, n = make_dataloader(3, None, batch_size=2, image_size=8,
                                num_classes=4)
    assert n == 20  # synthetic 10*batch
    xb, yb = next(iter(loader))
    assert xb.shape == (2, 3, 8, 8)


def test_dataloader_num_workers(tmp_path):
    from mpi4dl_amd.data import make_dataloader

    root, _, _ = _make_cifar(str(tmp_path), n=12)
    loader, n = make_dataloader(2, root, batch_size=4, image_size=32,
                                num_classes=10, num_workers=2)
    batches = list(loader)
    assert len(batches) == 3 and batches[0][0].shape == (4, 3, 32, 32)


def _runner_body(rank, world, root):
    import sys

    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "benchmarks"))
    from runner import run_training

    from mpi4dl_amd.parser import get_parser

    args = get_parser().parse_args([
        "--model", "resnet", "--app", "2", "--datapath", root,
        "--batch-size", "4", "--parts", "2", "--split-size", str(world),
        "--image-size", "32", "--num-layers", "9", "--num-filters", "4",
        "--num-epochs", "1", "--num-steps", "2", "--backend", "gloo",
        "--enable-evaluation",  # no test_batch on disk -> train-split fallback
    ])
    times = run_training(args, "lp")
    return len(times)


def test_runner_cifar_lp(tmp_path):
    # 16 samples so batch 4 with drop_last yields >= 2 batches
    root, _, _ = _make_cifar(str(tmp_path), n=16)
    got = run_distributed(_runner_body, 2, (root,))
    assert got[0] == 2
