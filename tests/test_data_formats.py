"""Data-layer tests against the real on-disk formats (generated fixtures):
CIFAR-10 pickle batches, ImageFolder trees, ImageNet-LT lists, and the
long-tail CIFAR subsampling."""

import os
import pickle

import numpy as np
import pytest
import torch


@pytest.fixture
def cifar_dir(tmp_path):
    root = tmp_path / "cifar"
    base = root / "cifar-10-batches-py"
    base.mkdir(parents=True)
    rng = np.random.RandomState(0)
    for i in range(1, 6):
        data = rng.randint(0, 255, size=(20, 3072), dtype=np.uint8)
        labels = rng.randint(0, 10, size=20).tolist()
        with open(base / f"data_batch_{i}", "wb") as fh:
            pickle.dump({"data": data, "labels": labels}, fh)
    data = rng.randint(0, 255, size=(30, 3072), dtype=np.uint8)
    with open(base / "test_batch", "wb") as fh:
        pickle.dump({"data": data, "labels": rng.randint(0, 10, 30).tolist()}, fh)
    return str(root)


def test_cifar10_pickle_format(cifar_dir):
    from active_learning_amd.data.cifar10 import get_data_cifar10
    train, test, al = get_data_cifar10(cifar_dir)
    assert len(train) == 100 and len(test) == 30 and len(al) == 100
    x, y, idx = train[7]
    assert x.shape == (3, 32, 32) and idx == 7 and 0 <= y < 10
    # al_set uses eval transforms: deterministic
    a1 = al[3][0]
    a2 = al[3][0]
    assert torch.equal(a1, a2)


def test_cifar10_missing_dir():
    from active_learning_amd.data.cifar10 import CustomCIFAR10
    with pytest.raises(FileNotFoundError):
        CustomCIFAR10("/nonexistent/path")


def test_imbalanced_cifar10(cifar_dir):
    from active_learning_amd.data.imbalanced_cifar10 import get_data_imbalanced_cifar10
    imb = {"imbalance_type": "exp", "imbalance_factor": 0.5, "imbalance_seed": 0}
    train, test, al = get_data_imbalanced_cifar10(cifar_dir, imbalance_args=imb)
    assert len(train) < 100          # subsampled long tail
    assert len(train) == len(al)     # same subsample (same seed)
    assert train.targets == al.targets
    counts = train.get_num_classes_list()
    assert counts[0] >= counts[-1]   # decaying per-class counts


def test_imagefolder(tmp_path):
    from PIL import Image
    from active_learning_amd.data.imagenet import CustomImageNet
    root = tmp_path / "train"
    for cls in ["n01", "n02"]:
        (root / cls).mkdir(parents=True)
        for i in range(3):
            Image.new("RGB", (40, 40), color=(i * 30, 0, 0)).save(
                root / cls / f"img{i}.jpg")
    from active_learning_amd.data.transforms import imagenet_transforms
    _, evalt = imagenet_transforms()
    ds = CustomImageNet(str(root), transform=evalt)
    assert len(ds) == 6
    x, y, idx = ds[4]
    assert x.shape == (3, 224, 224) and y in (0, 1) and idx == 4


def test_imagenet_lt_lists(tmp_path):
    from PIL import Image
    from active_learning_amd.data.imagenet import ImbalanceImagenet
    img_dir = tmp_path / "imgs"
    img_dir.mkdir()
    lines = []
    for i in range(4):
        p = f"imgs/im{i}.jpg"
        Image.new("RGB", (16, 16)).save(tmp_path / p)
        lines.append(f"{p} {i % 2}")
    list_file = tmp_path / "ImageNet_LT_train.txt"
    list_file.write_text("\n".join(lines))
    ds = ImbalanceImagenet(str(tmp_path), str(list_file))
    assert len(ds) == 4
    x, y, idx = ds[2]
    assert y == 0 and idx == 2


def test_imagenet_lt_fixture_e2e(tmp_path):
    """Committed ImageNet-LT fixture (tests/fixtures/imagenet_lt — the
    reference's exact `path label` line format,
    src/data_utils/ImageNet_LT/ImageNet_LT_test.txt:1) through a full
    imbalanced-imagenet debug round."""
    import os
    import shutil
    from PIL import Image
    from active_learning_amd.cli import get_args
    from active_learning_amd.main_al import main

    fix = os.path.join(os.path.dirname(__file__), "fixtures", "imagenet_lt")
    for name in ("ImageNet_LT_train.txt", "ImageNet_LT_test.txt"):
        shutil.copy(os.path.join(fix, name), tmp_path / name)
        with open(os.path.join(fix, name)) as fh:
            for line in fh:
                rel, lbl = line.split()
                assert rel.split("/")[0] in ("train", "val")
                p = tmp_path / rel
                p.parent.mkdir(parents=True, exist_ok=True)
                Image.new("RGB", (24, 24), (int(lbl) * 50, 10, 10)).save(p)

    args = get_args([
        "--dataset", "imbalanced_imagenet", "--dataset_dir", str(tmp_path),
        "--strategy", "RandomSampler", "--rounds", "1", "--round_budget", "5",
        "--n_epoch", "1", "--early_stop_patience", "1", "--debug_mode",
        "--ckpt_path", str(tmp_path / "ck"), "--log_dir", str(tmp_path / "lg"),
        "--model", "SSLResNet18"])
    s = main(args)
    assert s.idxs_lb.sum() == 5
    assert len(s.train_set) == 50  # fixture train list length (debug cap == len)


def test_imbalanced_e2e_round(tmp_path, cifar_dir):
    """Imbalanced dataset + inverse-frequency weighted CE through a debug
    round (reference: imbalanced_training flag, strategy.py:352-356)."""
    from active_learning_amd.cli import get_args
    from active_learning_amd.main_al import main
    args = get_args([
        "--dataset", "imbalanced_cifar10", "--dataset_dir", cifar_dir,
        "--imbalance_type", "exp", "--imbalance_factor", "0.5",
        "--strategy", "RandomSampler", "--rounds", "1", "--round_budget", "5",
        "--n_epoch", "1", "--early_stop_patience", "1", "--debug_mode",
        "--ckpt_path", str(tmp_path / "ck"), "--log_dir", str(tmp_path / "lg"),
        "--model", "SSLResNet18"])
    s = main(args)
    assert s.imbalanced_training
    assert s.idxs_lb.sum() == 5
    w = s.generate_imbalanced_training_weights()
    assert abs(w.sum().item() - 1.0) < 1e-5


def test_synthetic_dataset_with_loader_workers():
    """Datasets must survive DataLoader worker processes (the real arg pools
    use num_workers=12; synthetic sets must pickle into workers too)."""
    import torch
    from active_learning_amd.data.synthetic import get_data_synthetic

    train, test, al = get_data_synthetic(10, 64, 16, (3, 8, 8), seed=3)
    loader = torch.utils.data.DataLoader(train, batch_size=16, num_workers=2,
                                         persistent_workers=False)
    seen = 0
    for x, y, idx in loader:
        seen += x.shape[0]
        assert x.shape[1:] == (3, 8, 8)
    assert seen == 64
