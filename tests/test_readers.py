"""Real dataset I/O: tiny on-disk fixtures exercising the npy-index ImageNet
reader and the block-offset ImageNet22k tarball reader end to end."""

import gzip
import io
import os

import numpy as np
import pytest
import torch

from dinov3_amd.data import make_dataset
from dinov3_amd.data.readers import (
    ImageNet22kTarballReader,
    ImageNetIndexReader,
    decode_image_bytes,
    dump_imagenet22k_index,
    dump_imagenet_index,
    make_class_tarball,
    scan_tarball_blocks,
    write_blocks_log,
)


def _jpeg_bytes(color, size=32):
    from PIL import Image

    im = Image.new("RGB", (size, size), color)
    buf = io.BytesIO()
    im.save(buf, format="JPEG", quality=95)
    return buf.getvalue()


COLORS = [(255, 0, 0), (0, 255, 0), (0, 0, 255), (200, 200, 0)]


def _check_color(tensor, color, tol=0.08):
    mean = tensor.mean(dim=(1, 2))
    expect = torch.tensor(color, dtype=torch.float32) / 255.0
    assert (mean - expect).abs().max() < tol, (mean, expect)


# ----------------------------- ImageNet (npy index) -----------------------------


@pytest.fixture()
def imagenet_tree(tmp_path):
    root = tmp_path / "in1k"
    extra = tmp_path / "in1k-extra"
    classes = ["n01440764", "n01443537"]
    for ci, cid in enumerate(classes):
        d = root / "train" / cid
        d.mkdir(parents=True)
        for j in range(2):
            (d / f"{cid}_{10 + j}.JPEG").write_bytes(_jpeg_bytes(COLORS[ci * 2 + j]))
    n = dump_imagenet_index(str(root), str(extra), "train")
    assert n == 4
    return str(root), str(extra)


def test_imagenet_index_reader_roundtrip(imagenet_tree):
    root, extra = imagenet_tree
    r = ImageNetIndexReader(root, extra, "train")
    assert len(r) == 4
    targets = r.get_targets()
    assert sorted(targets.tolist()) == [0, 0, 1, 1]
    for i in range(4):
        img = decode_image_bytes(r.get_image_data(i))
        assert img.shape == (3, 32, 32)
        ci, j = divmod(i, 2)
        _check_color(img, COLORS[ci * 2 + j])
        assert r.get_target(i) == ci


def test_imagenet_dataset_string_real_mode(imagenet_tree):
    root, extra = imagenet_tree
    ds = make_dataset(dataset_str=f"ImageNet:split=TRAIN:root={root}:extra={extra}")
    assert len(ds) == 4
    img, target = ds[0]
    assert isinstance(img, torch.Tensor) and img.shape == (3, 32, 32)
    assert target == 0
    # entry dtype matches the reference contract (actual_index/class_index/
    # class_id/class_name) so Meta-built indexes interoperate
    entries = np.load(os.path.join(extra, "entries-TRAIN.npy"))
    assert set(entries.dtype.names) == {"actual_index", "class_index", "class_id", "class_name"}


def test_imagenet_dataset_synthetic_without_root():
    ds = make_dataset(dataset_str="ImageNet:split=VAL:length=8")
    assert len(ds) == 8
    img, target = ds[3]
    assert img.shape == (3, 224, 224)
    assert 0 <= target < 1000


# --------------------------- ImageNet22k (tarballs) ---------------------------


@pytest.fixture()
def in22k_tree(tmp_path):
    root = tmp_path / "in22k"
    extra = tmp_path / "in22k-extra"
    root.mkdir()
    classes = ["n02119789", "n02100735"]
    expected = {}
    for ci, cid in enumerate(sorted(classes)):
        members = []
        for j in range(2):
            data = _jpeg_bytes(COLORS[ci * 2 + j])
            name = f"{cid}_{j}.JPEG"
            if ci == 1 and j == 1:  # one gzip member, like the 26 gzipped 22k samples
                data_on_disk = gzip.compress(data)
            else:
                data_on_disk = data
            members.append((name, data_on_disk))
            expected[(ci, j)] = COLORS[ci * 2 + j]
        make_class_tarball(str(root / f"{cid}.tar"), members)
        write_blocks_log(str(root / f"{cid}.tar"), str(root / "blocks" / f"{cid}.log"))
    return str(root), str(extra), expected


def test_scan_tarball_blocks_matches_blocks_log(in22k_tree):
    root, _, _ = in22k_tree
    for cid in ("n02100735", "n02119789"):
        from dinov3_amd.data.readers import parse_blocks_log

        scanned = list(scan_tarball_blocks(os.path.join(root, f"{cid}.tar")))
        logged = list(parse_blocks_log(os.path.join(root, "blocks", f"{cid}.log")))
        assert logged[-1][0] == "** Block of NULs **"
        assert [(n, s) for n, s, _ in scanned] == logged[:-1]
        # consecutive members: each end == next start; last end == NULs marker
        for (_, _, e1), (_, s2) in zip(scanned, logged[1:]):
            assert e1 == s2


@pytest.mark.parametrize("use_logs", [True, False])
def test_imagenet22k_reader_roundtrip(in22k_tree, use_logs):
    root, extra, expected = in22k_tree
    n = dump_imagenet22k_index(root, extra, use_blocks_logs=use_logs)
    assert n == 4
    r = ImageNet22kTarballReader(root, extra)
    assert len(r) == 4
    for i in range(4):
        ci = r.get_target(i)
        img = decode_image_bytes(r.get_image_data(i))
        j = i % 2
        _check_color(img, expected[(ci, j)])


def test_imagenet22k_dataset_string_real_mode(in22k_tree):
    root, extra, _ = in22k_tree
    dump_imagenet22k_index(root, extra)
    ds = make_dataset(dataset_str=f"ImageNet22k:root={root}:extra={extra}")
    assert len(ds) == 4
    img, target = ds[1]
    assert img.shape == (3, 32, 32)
    assert target in (0, 1)


def test_imagenet22k_with_multicrop_transform(in22k_tree):
    """The real reader feeds the standard DINO multi-crop pipeline."""
    from dinov3_amd.data import DataAugmentationDINO

    root, extra, _ = in22k_tree
    dump_imagenet22k_index(root, extra)
    aug = DataAugmentationDINO((0.32, 1.0), (0.05, 0.32), 2,
                               global_crops_size=32, local_crops_size=16)
    ds = make_dataset(dataset_str=f"ImageNet22k:root={root}:extra={extra}", transform=aug)
    out, _ = ds[0]
    assert len(out["global_crops"]) == 2
    assert out["global_crops"][0].shape == (3, 32, 32)
    assert len(out["local_crops"]) == 2


# ------------------------------- ADE20K / Coco -------------------------------


def test_ade20k_reader_and_dataset(tmp_path):
    from PIL import Image

    root = tmp_path / "ade"
    (root / "images").mkdir(parents=True)
    (root / "annotations").mkdir()
    names = ["training/a_0001.jpg", "training/a_0002.jpg"]
    for i, n in enumerate(names):
        (root / "images" / n).parent.mkdir(parents=True, exist_ok=True)
        (root / "images" / n).write_bytes(_jpeg_bytes(COLORS[i]))
        mask = Image.fromarray(np.full((32, 32), i + 1, dtype=np.uint8))
        seg = root / "annotations" / (os.path.splitext(n)[0] + ".png")
        seg.parent.mkdir(parents=True, exist_ok=True)
        mask.save(seg)
    (root / "ADE20K_object150_train.txt").write_text("\n".join(names))

    from dinov3_amd.data.readers import ADE20KReader

    r = ADE20KReader(str(root), "train")
    assert len(r) == 2
    img = decode_image_bytes(r.get_image_data(0))
    _check_color(img, COLORS[0])
    t = r.get_target(1)
    assert t.shape == (32, 32) and int(t[0, 0]) == 2

    ds = make_dataset(dataset_str=f"ADE20K:split=TRAIN:root={root}")
    img2, t2 = ds[0]
    assert img2.shape == (3, 32, 32)
    assert t2.shape == (32, 32)


def test_coco_captions_reader_and_dataset(tmp_path):
    import json

    root = tmp_path / "coco"
    img_dir = root / "train2014" / "train2014"
    img_dir.mkdir(parents=True)
    ann_dir = root / "annotations_trainval2014" / "annotations"
    ann_dir.mkdir(parents=True)
    (img_dir / "img1.jpg").write_bytes(_jpeg_bytes(COLORS[0]))
    (img_dir / "img2.jpg").write_bytes(_jpeg_bytes(COLORS[1]))
    ann = {
        "images": [{"id": 10, "file_name": "img1.jpg"}, {"id": 20, "file_name": "img2.jpg"}],
        "annotations": [
            {"image_id": 10, "caption": "a red square"},
            {"image_id": 10, "caption": "very red"},
            {"image_id": 20, "caption": "a green square"},
        ],
    }
    (ann_dir / "captions_train2014.json").write_text(json.dumps(ann))

    from dinov3_amd.data.readers import CocoCaptionsReader

    r = CocoCaptionsReader(str(root), "train")
    assert len(r) == 2
    _check_color(decode_image_bytes(r.get_image_data(0)), COLORS[0])
    assert "red" in r.get_target(0)
    assert r.get_target(1) == "a green square"

    ds = make_dataset(dataset_str=f"CocoCaptions:split=TRAIN:root={root}")
    img, cap = ds[1]
    assert img.shape == (3, 32, 32)
    assert isinstance(cap, str) and "green" in cap
