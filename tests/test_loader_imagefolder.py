"""ImageFolderDataset: real-image decode path (PIL)."""

import os

import pytest
import torch

PIL = pytest.importorskip("PIL")
from PIL import Image  # noqa: E402

from moco_amd.loader import ImageFolderDataset, build_dataset  # noqa: E402


@pytest.fixture
def image_tree(tmp_path):
    for cls, size in [("cat", (100, 80)), ("dog", (60, 140))]:
        d = tmp_path / "train" / cls
        d.mkdir(parents=True)
        for i in range(3):
            img = Image.new("RGB", size, color=(i * 40, 100, 200))
            img.save(str(d / f"{i}.png"))
    return str(tmp_path)


def test_imagefolder_fixed_shapes_and_labels(image_tree):
    ds = ImageFolderDataset(os.path.join(image_tree, "train"), load_size=48)
    assert len(ds) == 6
    assert ds.class_to_idx == {"cat": 0, "dog": 1}
    shapes = set()
    labels = []
    for i in range(len(ds)):
        img, label = ds[i]
        shapes.add(tuple(img.shape))
        labels.append(label)
        assert 0.0 <= img.min() and img.max() <= 1.0
    assert shapes == {(3, 48, 48)}  # square, batchable
    assert sorted(set(labels)) == [0, 1]


def test_imagefolder_batches_with_dataloader(image_tree):
    ds = build_dataset(image_tree, load_size=32)
    loader = torch.utils.data.DataLoader(ds, batch_size=4, shuffle=False)
    raw, labels = next(iter(loader))
    assert raw.shape == (4, 3, 32, 32)
    assert labels.shape == (4,)
