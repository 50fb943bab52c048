"""Datasets.

Capability parity with reference dfd/timm/data/dataset.py:
  * `Dataset` — ImageFolder-style single-image dataset (:77)
  * `DeepFakeDataset_v3` — the ACTIVE video-frame-group dataset (:378-528):
    reads `real_list.txt`/`fake_list.txt` (`name:img_num` lines), seeded
    train/val split where val is the complement of the train sample,
    label-balance by splitting fakes into len(real) groups with per-group
    round-robin, 4 JPEG frames per item padded by repeating frame 0,
    fake=label 0 / real=label 1, optional `noise_fake` label flipping.

MI355X-native fix (SURVEY.md §5 race hazards): the reference mutates a
`fakeIndexes` cursor inside `__getitem__` (dataset.py:486-491), which under
num_workers>0 gives each worker its own copy — non-deterministic round-robin.
Here the round-robin position is derived from an explicit epoch counter
(`set_epoch`, called by the trainer next to sampler.set_epoch), so every
worker sees the same cursor and runs are reproducible.
"""

import os
import random

import numpy as np
import torch.utils.data as data

try:
    from PIL import Image
except ImportError:  # PIL is present in this image; guard for minimal envs
    Image = None

IMG_EXTENSIONS = [".png", ".jpg", ".jpeg"]


def natural_key(string_):
    import re

    return [int(s) if s.isdigit() else s for s in re.split(r"(\d+)", string_.lower())]


def find_images_and_targets(folder, types=IMG_EXTENSIONS, class_to_idx=None,
                            leaf_name_only=True, sort=True):
    labels = []
    filenames = []
    for root, _, files in os.walk(folder, topdown=False):
        rel_path = os.path.relpath(root, folder) if (root != folder) else ""
        label = os.path.basename(rel_path) if leaf_name_only else rel_path.replace(os.path.sep, "_")
        for f in files:
            base, ext = os.path.splitext(f)
            if ext.lower() in types:
                filenames.append(os.path.join(root, f))
                labels.append(label)
    if class_to_idx is None:
        unique_labels = set(labels)
        sorted_labels = list(sorted(unique_labels, key=natural_key))
        class_to_idx = {c: idx for idx, c in enumerate(sorted_labels)}
    images_and_targets = [
        (f, class_to_idx[l]) for f, l in zip(filenames, labels) if l in class_to_idx
    ]
    if sort:
        images_and_targets = sorted(images_and_targets, key=lambda k: natural_key(k[0]))
    return images_and_targets, class_to_idx


def load_class_map(class_names):
    """'fake,real' -> {'fake': 0, 'real': 1} (reference load_class_map_v2)."""
    if isinstance(class_names, str):
        class_names = class_names.split(",")
    return {name: idx for idx, name in enumerate(class_names)}


class Dataset(data.Dataset):
    """ImageFolder-style dataset (reference dataset.py:77)."""

    def __init__(self, root, load_bytes=False, transform=None, class_to_idx=None):
        images, class_to_idx = find_images_and_targets(root, class_to_idx=class_to_idx)
        if len(images) == 0:
            raise RuntimeError(f"Found 0 images in subfolders of {root}")
        self.root = root
        self.samples = images
        self.imgs = self.samples
        self.class_to_idx = class_to_idx
        self.load_bytes = load_bytes
        self.transform = transform

    def __getitem__(self, index):
        path, target = self.samples[index]
        img = open(path, "rb").read() if self.load_bytes else Image.open(path).convert("RGB")
        if self.transform is not None:
            img = self.transform(img)
        if target is None:
            target = -1
        return img, target

    def __len__(self):
        return len(self.samples)

    def filenames(self, indices=[], basename=False):
        if indices:
            if basename:
                return [os.path.basename(self.samples[i][0]) for i in indices]
            return [self.samples[i][0] for i in indices]
        if basename:
            return [os.path.basename(x[0]) for x in self.samples]
        return [x[0] for x in self.samples]

    def set_transform(self, transform):
        self.transform = transform


def get_all_images_list_v3(list_files):
    """Parse `name:img_num` list files; each entry also remembers which root
    it came from (reference dataset.py:362-373)."""
    files = []
    for root_index, list_file in enumerate(list_files):
        if not os.path.isfile(list_file):
            continue
        with open(list_file, "r") as f:
            for line in f.readlines():
                line = line.strip()
                if not line:
                    continue
                name, img_num = line.split(":")
                files.append((name, int(img_num), root_index))
    return files


class DeepFakeDataset_v3(data.Dataset):
    """Video-frame-group deepfake dataset — see module docstring.

    Item layout: 4 RGB frames from `<root>/{fake,real}/<name>/{0..3}.jpg`;
    when a clip has img_num<4 frames, frame 0 is repeated in front
    (reference dataset.py:496-512). Labels: fake=0, real=1.
    """

    def __init__(self, roots, class_names, load_bytes=False, transform=None,
                 transform_rotateds=None, frac=1, n=None, random_state=None,
                 train_split=False, train_ratio=0.0, is_training=False,
                 label_balance=False, noise_fake=False):
        class_to_idx = load_class_map(class_names)
        if isinstance(roots, str):
            roots = [roots]
        self.roots = roots
        self.noise_fake = noise_fake
        self.class_to_idx = class_to_idx
        self.load_bytes = load_bytes
        self.transform = transform
        self.transform_rotateds = transform_rotateds
        self._epoch = 0

        real_listfiles = [os.path.join(root, "real_list.txt") for root in roots]
        fake_listfiles = [os.path.join(root, "fake_list.txt") for root in roots]
        self.real_images = get_all_images_list_v3(real_listfiles)
        self.fake_images = get_all_images_list_v3(fake_listfiles)

        if train_split:
            # deterministic split: val = complement of the seeded train sample
            # (reference dataset.py:424-438)
            random.seed(random_state if random_state is not None else 0)
            sample_real_images = self.real_images
            sample_fake_images = self.fake_images
            if int(len(self.real_images) * train_ratio) >= 1:
                sample_real_images = random.sample(
                    self.real_images, int(len(self.real_images) * train_ratio))
            if int(len(self.fake_images) * train_ratio) >= 1:
                sample_fake_images = random.sample(
                    self.fake_images, int(len(self.fake_images) * train_ratio))
            if is_training:
                self.real_images = sample_real_images
                self.fake_images = sample_fake_images
            else:
                self.real_images = list(set(self.real_images) - set(sample_real_images))
                self.fake_images = list(set(self.fake_images) - set(sample_fake_images))
        else:
            if 0 < frac < 1:
                if random_state is not None:
                    random.seed(random_state)
                if int(len(self.real_images) * frac) >= 1:
                    self.real_images = random.sample(self.real_images, int(len(self.real_images) * frac))
                if int(len(self.fake_images) * frac) >= 1:
                    self.fake_images = random.sample(self.fake_images, int(len(self.fake_images) * frac))
            elif n:
                if random_state is not None:
                    random.seed(random_state)
                if len(self.real_images) > n:
                    self.real_images = random.sample(self.real_images, n)
                if len(self.fake_images) > n:
                    self.fake_images = random.sample(self.fake_images, n)

        # label balancing: split fakes into len(real) groups; each epoch the
        # group contributes its next member round-robin
        # (reference dataset.py:460-491)
        if len(self.fake_images) > 0:
            if label_balance:
                if len(self.real_images) == 0 or len(self.real_images) > len(self.fake_images):
                    self.fake_images = np.array_split(self.fake_images, len(self.fake_images))
                else:
                    self.fake_images = np.array_split(self.fake_images, len(self.real_images))
            else:
                self.fake_images = np.array_split(self.fake_images, len(self.fake_images))

    def set_epoch(self, epoch):
        """Advance the deterministic round-robin cursor (replaces the
        reference's per-worker mutable `fakeIndexes`, dataset.py:486-491)."""
        self._epoch = int(epoch)

    def _frame_paths(self, cls, name, img_num, root_index):
        to_load_num = 4 - img_num
        paths = [os.path.join(self.roots[root_index], cls, name, "0.jpg")] * to_load_num
        paths += [
            os.path.join(self.roots[root_index], cls, name, f"{i}.jpg")
            for i in range(img_num)
        ]
        return paths

    def __getitem__(self, index):
        if index < len(self.fake_images):
            target = 0
            group = self.fake_images[index]
            fake_index = self._epoch % len(group)
            img_name, img_num, root_index = group[fake_index]
            img_paths = self._frame_paths("fake", str(img_name), int(img_num), int(root_index))
        else:
            target = 1
            img_name, img_num, root_index = self.real_images[index - len(self.fake_images)]
            img_paths = self._frame_paths("real", str(img_name), int(img_num), int(root_index))

        imgs = [
            open(p, "rb").read() if self.load_bytes else Image.open(p).convert("RGB")
            for p in img_paths
        ]
        if self.transform is not None:
            imgs = self.transform(imgs)
        if target == 0 and self.noise_fake:
            target = 0 if random.random() < 0.5 else 1
        return imgs, target

    def __len__(self):
        return len(self.real_images) + len(self.fake_images)

    def set_transform(self, transform):
        self.transform = transform


class DeepFakeDataset_v1(data.Dataset):
    """Legacy pair-file dataset (reference dataset.py:531): `fake:real:rotated`
    lines; even indices yield the fake image (label 0), odd the real (1)."""

    def __init__(self, root, result_file, load_bytes=False, transform=None,
                 transform_rotateds=None, class_map=""):
        self.root = root
        self.load_bytes = load_bytes
        self.transform = transform
        self.transform_rotateds = transform_rotateds
        self.results = []
        with open(result_file, "r") as f:
            for line in f.readlines():
                line_s = line.strip().split(":")
                if len(line_s) != 3:
                    continue
                self.results.append((line_s[0], line_s[1], int(line_s[2])))
        if len(self.results) == 0:
            raise RuntimeError(f"Found 0 entries in {result_file}")

    def __getitem__(self, index):
        target = index % 2
        result_index = index // 2
        img_path = self.results[result_index][target]
        rotated = self.results[result_index][2]
        img = open(img_path, "rb").read() if self.load_bytes else Image.open(img_path).convert("RGB")
        if self.transform_rotateds is not None:
            img = self.transform_rotateds[rotated](img)
        if self.transform is not None:
            img = self.transform(img)
        return img, target

    def __len__(self):
        return 2 * len(self.results)

    def set_transform(self, transform, transform_rotateds=None):
        self.transform = transform
        if transform_rotateds is not None:
            self.transform_rotateds = transform_rotateds


class DeepFakeDataset_v1_bak(data.Dataset):
    """Paired legacy variant (reference dataset.py:126): each item returns
    (fake_img, real_img, fake_rotated, real_rotated); real list shuffled at
    construction so pairs are decorrelated."""

    def __init__(self, root, result_file, load_bytes=False, transform=None,
                 transform_rotateds=None, class_map=""):
        self.root = root
        self.load_bytes = load_bytes
        self.transform = transform
        self.transform_rotateds = transform_rotateds
        self.results_fake = []
        self.results_real = []
        with open(result_file, "r") as f:
            for line in f.readlines():
                line_s = line.strip().split(":")
                if len(line_s) != 3:
                    continue
                self.results_fake.append((line_s[0], int(line_s[2])))
                self.results_real.append((line_s[1], int(line_s[2])))
        if not self.results_fake:
            raise RuntimeError(f"Found 0 entries in {result_file}")
        random.shuffle(self.results_real)

    def _load(self, path):
        return open(path, "rb").read() if self.load_bytes else Image.open(path).convert("RGB")

    def __getitem__(self, index):
        fake_path, fake_rot = self.results_fake[index]
        real_path, real_rot = self.results_real[index]
        fake_img, real_img = self._load(fake_path), self._load(real_path)
        if self.transform_rotateds is not None:
            fake_img = self.transform_rotateds[fake_rot](fake_img)
            real_img = self.transform_rotateds[real_rot](real_img)
        if self.transform is not None:
            fake_img = self.transform(fake_img)
            real_img = self.transform(real_img)
        return fake_img, real_img, fake_rot, real_rot

    def __len__(self):
        return len(self.results_fake)


def get_all_images_list(list_files):
    """Read `<name>` lines from per-root list files -> [(name, root_idx)]
    (reference dataset.py get_all_images_list)."""
    files = []
    for root_index, list_file in enumerate(list_files):
        if not os.path.isfile(list_file):
            continue
        with open(list_file, "r") as f:
            files += [(line.strip(), root_index) for line in f.readlines()]
    return files


class DeepFakeDataset_v2(data.Dataset):
    """Single-image predecessor of v3 (reference dataset.py:284): per-root
    real_list/fake_list files, fakes split into len(real) groups with random
    choice per group; even index = fake (0), odd = real (1); rotation id
    parsed from the file-name suffix `*_<rot>.jpg`."""

    def __init__(self, roots, class_names, load_bytes=False, transform=None,
                 transform_rotateds=None, frac=1):
        if isinstance(roots, str):
            roots = [roots]
        self.roots = roots
        self.class_to_idx = load_class_map(class_names)
        self.load_bytes = load_bytes
        self.transform = transform
        self.transform_rotateds = transform_rotateds

        real_lists = [os.path.join(r, "real_list.txt") for r in roots]
        fake_lists = [os.path.join(r, "fake_list.txt") for r in roots]
        self.real_images = get_all_images_list(real_lists)
        fake_images = get_all_images_list(fake_lists)
        assert len(fake_images) >= len(self.real_images)
        assert len(self.real_images) > 0
        if 0 < frac < 1:
            random.seed(1024)
            self.real_images = random.sample(self.real_images, int(len(self.real_images) * frac))
            fake_images = random.sample(fake_images, int(len(fake_images) * frac))
        self.fake_images = np.array_split(np.array(fake_images, dtype=object), len(self.real_images))

    def __getitem__(self, index):
        target = index % 2
        ti = index // 2
        if target == 0:
            img_name, root_index = random.choice(list(self.fake_images[ti]))
            img_path = os.path.join(self.roots[int(root_index)], "fake", img_name)
        else:
            img_name, root_index = self.real_images[ti]
            img_path = os.path.join(self.roots[int(root_index)], "real", img_name)
        rotated = int(img_path.split("_")[-1].split(".")[0])
        img = open(img_path, "rb").read() if self.load_bytes else Image.open(img_path).convert("RGB")
        if self.transform_rotateds is not None:
            img = self.transform_rotateds[rotated](img)
        if self.transform is not None:
            img = self.transform(img)
        return img, target

    def __len__(self):
        return 2 * len(self.real_images)

    def set_transform(self, transform, transform_rotateds=None):
        self.transform = transform
        if transform_rotateds is not None:
            self.transform_rotateds = transform_rotateds


class ConcatDataset(data.ConcatDataset):
    """Concatenation with transform pass-through (reference dataset.py:229)."""

    def set_transform(self, transform):
        for ds in self.datasets:
            if hasattr(ds, "set_transform"):
                ds.set_transform(transform)
            else:
                ds.transform = transform


class DatasetTar(data.Dataset):
    """Tar-archive image dataset (reference dataset.py:602): folder name
    inside the archive is the class label; the tar handle is reopened
    lazily per worker process."""

    def __init__(self, root, load_bytes=False, transform=None, class_map=""):
        import os as _os
        import tarfile

        assert _os.path.isfile(root)
        self.root = root
        class_to_idx = load_class_map(class_map) if class_map else None
        with tarfile.open(root) as tf:
            files, labels = [], []
            for ti in tf.getmembers():
                if not ti.isfile():
                    continue
                dirname, basename = _os.path.split(ti.path)
                ext = _os.path.splitext(basename)[1]
                if ext.lower() in IMG_EXTENSIONS:
                    files.append(ti)
                    labels.append(_os.path.basename(dirname))
            if class_to_idx is None:
                sorted_labels = sorted(set(labels), key=natural_key)
                class_to_idx = {c: idx for idx, c in enumerate(sorted_labels)}
            self.samples = sorted(
                zip(files, [class_to_idx[l] for l in labels]),
                key=lambda k: natural_key(k[0].path))
            self.class_to_idx = class_to_idx
        self.tarfile = None  # lazy re-open per worker
        self.load_bytes = load_bytes
        self.transform = transform

    def __getitem__(self, index):
        import tarfile

        if self.tarfile is None:
            self.tarfile = tarfile.open(self.root)
        tarinfo, target = self.samples[index]
        iob = self.tarfile.extractfile(tarinfo)
        img = iob.read() if self.load_bytes else Image.open(iob).convert("RGB")
        if self.transform is not None:
            img = self.transform(img)
        return img, target

    def __len__(self):
        return len(self.samples)

    def set_transform(self, transform):
        self.transform = transform


class AugMixDataset(data.Dataset):
    """Clean + augmented split wrapper for AugMix JSD training
    (reference dataset.py:633): transform is a (base, augmentation,
    normalize) triple; item = (clean, aug1, ..., augN-1), label."""

    def __init__(self, dataset, num_splits=2):
        self.augmentation = None
        self.normalize = None
        self.dataset = dataset
        if self.dataset.transform is not None:
            self._set_transforms(self.dataset.transform)
        self.num_splits = num_splits

    def _set_transforms(self, x):
        assert isinstance(x, (list, tuple)) and len(x) == 3, \
            "Expecting a tuple/list of 3 transforms"
        self.dataset.transform = x[0]
        self.augmentation = x[1]
        self.normalize = x[2]

    @property
    def transform(self):
        return self.dataset.transform

    @transform.setter
    def transform(self, x):
        self._set_transforms(x)

    def _normalize(self, x):
        return x if self.normalize is None else self.normalize(x)

    def __getitem__(self, i):
        x, y = self.dataset[i]
        x_list = [self._normalize(x)]  # clean split
        for _ in range(self.num_splits - 1):
            x_list.append(self._normalize(self.augmentation(x)))
        return tuple(x_list), y

    def __len__(self):
        return len(self.dataset)


class SyntheticDeepFakeDataset(data.Dataset):
    """Synthetic face-crop dataset for benchmarking (no disk, no network):
    deterministic pseudo-random uint8 frame groups shaped like
    DeepFakeDataset_v3 output ((3*img_num, H, W) uint8 + binary label)."""

    def __init__(self, length=1024, img_size=600, img_num=4, seed=0):
        self.length = length
        self.img_size = img_size
        self.img_num = img_num
        self.seed = seed

    def __getitem__(self, index):
        rng = np.random.RandomState((self.seed * 1_000_003 + index) % (2 ** 31))
        arr = rng.randint(0, 256, size=(3 * self.img_num, self.img_size, self.img_size), dtype=np.uint8)
        return arr, int(index % 2)

    def __len__(self):
        return self.length
