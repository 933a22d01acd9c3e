"""Real dataset I/O: ImageNet npy-index reader and ImageNet22k tarball reader.

The reference carries these readers but bypasses them with synthetic decode
(dinov3_jax/data/datasets/image_net.py:27-337, image_net_22k.py). Here the
on-disk contracts are honored for real so existing index files interoperate:

* ImageNet: an "extra" directory with ``entries-<SPLIT>.npy`` (structured
  array: actual_index <u4, class_index <u4, class_id U*, class_name U*),
  ``class-ids-<SPLIT>.npy`` and ``class-names-<SPLIT>.npy``; images under
  ``root/<split>/<class_id>/<class_id>_<actual_index>.JPEG`` for train and
  ``root/val/ILSVRC2012_val_<%08d>.JPEG`` for val/test.
* ImageNet22k: per-class GNU tarballs ``root/<class_id>.tar`` addressed by
  512-byte block offsets, with an ``entries.npy`` index (class_index <u4,
  class_id U*, start_offset <u4, end_offset <u4, filename U*) and
  ``class-ids.npy`` in the "extra" dir. Entries address whole tar members
  (header block included); a handful of members are gzip-compressed and are
  transparently decompressed. Tarballs are mmap'd with a small LRU cache so
  dataloader workers don't exhaust file descriptors.

Index builders (``dump_imagenet_index``/``dump_imagenet22k_index``) recreate
the npy files from a plain ImageFolder tree / a directory of tarballs, so the
tiny-fixture tests and first-time users don't need Meta's original index.
Decoding is PIL -> RGB -> uint8 CHW torch tensor (no torchvision dependency).
"""

from __future__ import annotations

import io
import logging
import mmap
import os
import tarfile
from functools import lru_cache
from gzip import GzipFile
from typing import Iterator, List, Optional, Tuple

import numpy as np
import torch

logger = logging.getLogger("dinov3")

TAR_BLOCK = 512


def decode_image_bytes(data: bytes) -> torch.Tensor:
    """JPEG/PNG bytes -> float32 CHW tensor in [0,1]."""
    from PIL import Image

    with Image.open(io.BytesIO(data)) as im:
        im = im.convert("RGB")
        arr = np.array(im, dtype=np.uint8)
    return torch.from_numpy(arr).permute(2, 0, 1).contiguous().float() / 255.0


# --------------------------------------------------------------------------
# ImageNet (npy index over an ImageFolder-style tree)
# --------------------------------------------------------------------------


def imagenet_entries_name(split: str) -> str:
    return f"entries-{split.upper()}.npy"


def imagenet_relpath(split: str, class_id: str, actual_index: int) -> str:
    """Image path relative to root (reference image_net.py:45-51)."""
    if split.lower() == "train":
        return os.path.join(split.lower(), class_id, f"{class_id}_{actual_index}.JPEG")
    return os.path.join(split.lower(), f"ILSVRC2012_{split.lower()}_{actual_index:08d}.JPEG")


class ImageNetIndexReader:
    """mmap'd npy index + per-image file reads."""

    def __init__(self, root: str, extra: str, split: str):
        self.root = root
        self.extra = extra
        self.split = split
        self.entries = np.load(os.path.join(extra, imagenet_entries_name(split)), mmap_mode="r")
        ids_path = os.path.join(extra, f"class-ids-{split.upper()}.npy")
        self.class_ids = np.load(ids_path, mmap_mode="r") if os.path.exists(ids_path) else None

    def __len__(self) -> int:
        return len(self.entries)

    def get_image_data(self, index: int) -> bytes:
        e = self.entries[index]
        rel = imagenet_relpath(self.split, str(e["class_id"]), int(e["actual_index"]))
        with open(os.path.join(self.root, rel), "rb") as f:
            return f.read()

    def get_target(self, index: int) -> int:
        return int(self.entries[index]["class_index"])

    def get_targets(self) -> np.ndarray:
        return np.asarray(self.entries["class_index"])


def dump_imagenet_index(root: str, extra: str, split: str = "train") -> int:
    """Scan ``root/<split>/<class_id>/*.JPEG`` and write the npy index files.
    Returns the number of entries written."""
    split_dir = os.path.join(root, split.lower())
    class_dirs = sorted(d for d in os.listdir(split_dir)
                        if os.path.isdir(os.path.join(split_dir, d)))
    rows: List[Tuple[int, int, str, str]] = []
    for class_index, class_id in enumerate(class_dirs):
        for fname in sorted(os.listdir(os.path.join(split_dir, class_id))):
            base, ext = os.path.splitext(fname)
            if ext.upper() != ".JPEG":
                continue
            actual_index = int(base.split("_")[-1])
            rows.append((actual_index, class_index, class_id, class_id))
    if not rows:
        raise RuntimeError(f"no images found under {split_dir}")
    id_len = max(len(r[2]) for r in rows)
    name_len = max(len(r[3]) for r in rows)
    dtype = np.dtype([("actual_index", "<u4"), ("class_index", "<u4"),
                      ("class_id", f"U{id_len}"), ("class_name", f"U{name_len}")])
    entries = np.array(rows, dtype=dtype)
    os.makedirs(extra, exist_ok=True)
    np.save(os.path.join(extra, imagenet_entries_name(split)), entries)
    np.save(os.path.join(extra, f"class-ids-{split.upper()}.npy"),
            np.array(class_dirs, dtype=f"U{id_len}"))
    np.save(os.path.join(extra, f"class-names-{split.upper()}.npy"),
            np.array(class_dirs, dtype=f"U{id_len}"))
    logger.info("wrote ImageNet index: %d entries, %d classes -> %s",
                len(entries), len(class_dirs), extra)
    return len(entries)


# --------------------------------------------------------------------------
# ImageNet22k (per-class tarballs addressed by 512-byte block offsets)
# --------------------------------------------------------------------------


def scan_tarball_blocks(path: str) -> Iterator[Tuple[str, int, int]]:
    """Yield (member_name, start_block, end_block) for every regular member,
    by walking the raw 512-byte tar headers (GNU/ustar). end_block is the
    first block after the member's data padding — i.e. the next header."""
    with open(path, "rb") as f:
        block = 0
        while True:
            f.seek(block * TAR_BLOCK)
            header = f.read(TAR_BLOCK)
            if len(header) < TAR_BLOCK or header == b"\0" * TAR_BLOCK:
                return
            name = header[:100].split(b"\0", 1)[0].decode("utf-8", "replace")
            size_field = header[124:136].split(b"\0", 1)[0].strip()
            size = int(size_field or b"0", 8)
            typeflag = header[156:157]
            data_blocks = (size + TAR_BLOCK - 1) // TAR_BLOCK
            end = block + 1 + data_blocks
            # longname/longlink (GNU 'L'/'K') headers prefix the real member
            if typeflag == b"L":
                f.seek((block + 1) * TAR_BLOCK)
                name = f.read(size).split(b"\0", 1)[0].decode("utf-8", "replace")
                f.seek(end * TAR_BLOCK)
                real_header = f.read(TAR_BLOCK)
                size = int(real_header[124:136].split(b"\0", 1)[0].strip() or b"0", 8)
                typeflag = real_header[156:157]
                data_blocks = (size + TAR_BLOCK - 1) // TAR_BLOCK
                real_end = end + 1 + data_blocks
                if typeflag in (b"0", b"\0"):
                    yield name, block, real_end
                block = real_end
                continue
            if typeflag in (b"0", b"\0") and size > 0:
                yield name, block, end
            block = end


def parse_blocks_log(path: str) -> Iterator[Tuple[str, int]]:
    """Parse a ``tar -tR``-style blocks log: ``block NNN: filename`` lines,
    ending with the '** Block of NULs **' marker (reference contract,
    image_net_22k.py:160-190)."""
    with open(path) as f:
        for line in f:
            line = line.rstrip("\n")
            if not line:
                continue
            block_part, _, filename = line.partition(":")
            yield filename[1:], int(block_part[6:])


class ImageNet22kTarballReader:
    GZIP_MAGIC = (0x1F, 0x8B)

    def __init__(self, root: str, extra: str, mmap_cache_size: int = 16):
        self.root = root
        self.extra = extra
        self.entries = np.load(os.path.join(extra, "entries.npy"), mmap_mode="r")
        ids_path = os.path.join(extra, "class-ids.npy")
        self.class_ids = np.load(ids_path, mmap_mode="r") if os.path.exists(ids_path) else None

        @lru_cache(maxsize=mmap_cache_size)
        def _mmap_tarball(class_id: str) -> mmap.mmap:
            with open(os.path.join(root, f"{class_id}.tar"), "rb") as f:
                return mmap.mmap(f.fileno(), 0, access=mmap.ACCESS_READ)

        self._mmap_tarball = _mmap_tarball

    def __len__(self) -> int:
        return len(self.entries)

    def get_image_data(self, index: int) -> bytes:
        e = self.entries[index]
        m = self._mmap_tarball(str(e["class_id"]))
        data = m[int(e["start_offset"]) + TAR_BLOCK: int(e["end_offset"])]
        if len(data) >= 2 and tuple(data[:2]) == self.GZIP_MAGIC:
            with GzipFile(fileobj=io.BytesIO(data)) as g:
                data = g.read()
        return data

    def get_target(self, index: int) -> int:
        return int(self.entries[index]["class_index"])

    def get_targets(self) -> np.ndarray:
        return np.asarray(self.entries["class_index"])


def dump_imagenet22k_index(root: str, extra: str,
                           use_blocks_logs: Optional[bool] = None) -> int:
    """Build entries.npy/class-ids.npy for a directory of per-class tarballs.
    Block offsets come from ``root/blocks/<class_id>.log`` when present (the
    reference's contract), else directly from the tar headers."""
    class_ids = sorted(os.path.splitext(f)[0] for f in os.listdir(root) if f.endswith(".tar"))
    if not class_ids:
        raise RuntimeError(f"no tarballs under {root}")
    rows: List[Tuple[int, str, int, int, str]] = []
    for class_index, class_id in enumerate(class_ids):
        log_path = os.path.join(root, "blocks", f"{class_id}.log")
        if use_blocks_logs or (use_blocks_logs is None and os.path.exists(log_path)):
            marks = list(parse_blocks_log(log_path))
            assert marks and marks[-1][0] == "** Block of NULs **", \
                f"blocks log {log_path} missing end marker"
            for (fname, start), (_, end) in zip(marks, marks[1:]):
                rows.append((class_index, class_id, start * TAR_BLOCK, end * TAR_BLOCK, fname))
        else:
            for fname, start, end in scan_tarball_blocks(os.path.join(root, f"{class_id}.tar")):
                rows.append((class_index, class_id, start * TAR_BLOCK, end * TAR_BLOCK, fname))
    id_len = max(len(r[1]) for r in rows)
    fn_len = max(len(r[4]) for r in rows)
    dtype = np.dtype([("class_index", "<u4"), ("class_id", f"U{id_len}"),
                      ("start_offset", "<u4"), ("end_offset", "<u4"),
                      ("filename", f"U{fn_len}")])
    entries = np.array(rows, dtype=dtype)
    os.makedirs(extra, exist_ok=True)
    np.save(os.path.join(extra, "entries.npy"), entries)
    np.save(os.path.join(extra, "class-ids.npy"), np.array(class_ids, dtype=f"U{id_len}"))
    logger.info("wrote ImageNet22k index: %d entries, %d classes -> %s",
                len(entries), len(class_ids), extra)
    return len(entries)


def write_blocks_log(tar_path: str, log_path: str) -> None:
    """Generate the ``tar -tR``-style blocks log for a tarball (test fixture
    helper; production data ships these logs alongside the tarballs)."""
    os.makedirs(os.path.dirname(log_path), exist_ok=True)
    last_end = 0
    with open(log_path, "w") as out:
        for name, start, end in scan_tarball_blocks(tar_path):
            out.write(f"block {start}: {name}\n")
            last_end = end
        out.write(f"block {last_end}: ** Block of NULs **\n")


def make_class_tarball(path: str, images: List[Tuple[str, bytes]]) -> None:
    """Write a USTAR tarball of (filename, bytes) members (fixture helper)."""
    with tarfile.open(path, "w", format=tarfile.USTAR_FORMAT) as tar:
        for name, data in images:
            info = tarfile.TarInfo(name=name)
            info.size = len(data)
            tar.addfile(info, io.BytesIO(data))


# --------------------------------------------------------------------------
# ADE20K (split txt listing + images/ + annotations/ PNG masks)
# --------------------------------------------------------------------------


class ADE20KReader:
    """`root/ADE20K_object150_<split>.txt` lists image file names; images live
    under `root/images/<name>`, dense labels under
    `root/annotations/<name>.png` (reference ade20k.py:32-102 contract)."""

    def __init__(self, root: str, split: str):
        self.root = root
        list_path = os.path.join(root, f"ADE20K_object150_{split.lower()}.txt")
        with open(list_path) as f:
            names = sorted(f.read().strip().split("\n"))
        self.image_paths = [os.path.join("images", n) for n in names]
        self.target_paths = [os.path.join("annotations", os.path.splitext(n)[0] + ".png")
                             for n in names]

    def __len__(self) -> int:
        return len(self.image_paths)

    def get_image_data(self, index: int) -> bytes:
        with open(os.path.join(self.root, self.image_paths[index]), "rb") as f:
            return f.read()

    def get_target(self, index: int) -> torch.Tensor:
        """Dense segmentation labels as a [H, W] long tensor."""
        from PIL import Image

        with Image.open(os.path.join(self.root, self.target_paths[index])) as im:
            return torch.from_numpy(np.array(im, dtype=np.int64))

    def get_targets(self):
        return None


# --------------------------------------------------------------------------
# COCO captions (annotation json + image dir)
# --------------------------------------------------------------------------


class CocoCaptionsReader:
    """COCO captions layout (reference coco_captions.py:28-54):
    train = annotations_trainval2014/annotations/captions_train2014.json +
    train2014/train2014; val = the 2017 equivalents. Target = a random
    caption of the image (reference :97-102)."""

    def __init__(self, root: str, split: str, rng=None):
        import json
        import random as _random

        self._rng = rng or _random
        if split.lower() == "train":
            ann = os.path.join(root, "annotations_trainval2014/annotations/captions_train2014.json")
            image_dir = os.path.join(root, "train2014/train2014")
        else:
            ann = os.path.join(root, "annotations_trainval2017/annotations/captions_train2017.json")
            image_dir = os.path.join(root, "val2017/val2017")
        with open(ann) as f:
            all_annotations = json.load(f)
        data = {}
        for item in all_annotations["images"]:
            data[item["id"]] = {"image": os.path.join(image_dir, item["file_name"]),
                                "captions": []}
        for item in all_annotations["annotations"]:
            data[item["image_id"]]["captions"].append(item["caption"])
        self.entries = list(data.values())

    def __len__(self) -> int:
        return len(self.entries)

    def get_image_data(self, index: int) -> bytes:
        with open(self.entries[index]["image"], "rb") as f:
            return f.read()

    def get_target(self, index: int) -> str:
        caps = self.entries[index]["captions"]
        return self._rng.choice(caps) if caps else ""

    def get_targets(self):
        return None
