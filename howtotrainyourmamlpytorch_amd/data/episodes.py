"""Few-shot task-episode dataset (L1).

Re-implements the behavioral contract of the reference's
``FewShotLearningDatasetParallel`` (``data.py:111-552``):

* class-folder scan -> JSON path cache; classes split into train/val/test
  either by ``train_val_test_split`` fractions over the sorted class list or
  by pre-split ``train/ val/ test/`` directories (``sets_are_pre_split``);
* an episode is a **pure function of an integer seed**: a
  ``np.random.RandomState(seed)`` picks ``num_classes_per_set`` classes, a
  rotation k in {0..3} per class (Omniglot augmentation), and
  ``num_samples_per_class + num_target_samples`` images per class; class
  labels are remapped to 0..N-1 (``data.py:478-524``);
* ``__getitem__(idx)`` uses ``seed = set_seed + idx`` (``data.py:544-549``)
  so the task stream is deterministic and resumable by seed arithmetic.

The returned tensors are ``support [N, S, c, h, w]``, ``target
[N, T, c, h, w]`` (float32 in [0,1] for Omniglot / normalized RGB
otherwise), int64 labels, and the episode seed.
"""

from __future__ import annotations

import concurrent.futures
import json
import os
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

IMAGENET_MEAN = np.array([0.485, 0.456, 0.406], dtype=np.float32)
IMAGENET_STD = np.array([0.229, 0.224, 0.225], dtype=np.float32)

_IMAGE_EXTS = (".png", ".jpg", ".jpeg", ".bmp", ".gif")


def _class_key(path: str, indexes: Sequence[int]) -> str:
    parts = os.path.normpath(path).split(os.sep)
    return os.sep.join(parts[i] for i in indexes)


def scan_class_folders(root: str, indexes_of_folders_indicating_class: Sequence[int]
                       ) -> Dict[str, List[str]]:
    """Walk ``root`` and group image files by class key (reference:
    ``data.py:302-334``)."""
    classes: Dict[str, List[str]] = {}
    for dirpath, _dirnames, filenames in os.walk(root, followlinks=True):
        for fname in sorted(filenames):
            if not fname.lower().endswith(_IMAGE_EXTS):
                continue
            fpath = os.path.join(dirpath, fname)
            key = _class_key(fpath, indexes_of_folders_indicating_class)
            classes.setdefault(key, []).append(fpath)
    for v in classes.values():
        v.sort()
    return classes


def split_classes(classes: Dict[str, List[str]], fractions: Sequence[float],
                  ) -> Dict[str, Dict[str, List[str]]]:
    """Deterministic class-level split over the sorted class list."""
    names = sorted(classes.keys())
    n = len(names)
    n_train = int(np.floor(fractions[0] * n))
    n_val = int(np.floor(fractions[1] * n))
    sets = {
        "train": names[:n_train],
        "val": names[n_train:n_train + n_val],
        "test": names[n_train + n_val:],
    }
    return {s: {c: classes[c] for c in cs} for s, cs in sets.items()}


class FewShotEpisodeDataset(torch.utils.data.Dataset):
    def __init__(self, args, current_set: str = "train"):
        self.args = args
        self.num_classes_per_set = args.num_classes_per_set
        self.num_samples_per_class = args.num_samples_per_class
        self.num_target_samples = args.num_target_samples
        self.image_height = args.image_height
        self.image_width = args.image_width
        self.image_channels = args.image_channels
        self.augment_images = False
        self.current_set = current_set
        self.dataset_name = args.dataset_name

        self._npz_images = None  # [num_classes, spc, H, W] uint8 when npz-backed
        self._npz_class_index: Dict[str, int] = {}
        self.datasets = self._load_datapaths()
        # seeds per set: val/test share the fixed val seed (reference uses
        # the same seed for val and test streams, data.py:141-142)
        self.init_seed = {"train": args.train_seed, "val": args.val_seed,
                          "test": args.val_seed}
        self.seed = dict(self.init_seed)

        self._memory: Optional[Dict[str, List[np.ndarray]]] = None
        if getattr(args, "load_into_memory", False) and self._npz_images is None:
            self._preload()

    # ------------------------------------------------------------------
    def _cache_path(self) -> str:
        root = os.path.abspath(self.args.dataset_path)
        return root.rstrip(os.sep) + "_path_cache.json"

    def _load_datapaths(self) -> Dict[str, Dict[str, List[str]]]:
        cache = self._cache_path()
        reset = getattr(self.args, "reset_stored_paths", False) or \
            getattr(self.args, "reset_stored_filepaths", False)
        if os.path.isfile(cache) and not reset:
            try:
                with open(cache) as f:
                    return json.load(f)
            except (json.JSONDecodeError, OSError):
                pass
        root = self.args.dataset_path
        npz_path = None
        if str(root).endswith(".npz"):
            npz_path = root
        elif getattr(self.args, "load_from_npz_files", False):
            cand = str(root).rstrip(os.sep) + ".npz"
            if os.path.isfile(cand):
                npz_path = cand
        if npz_path is not None:
            return self._load_npz(npz_path)
        if not os.path.isdir(root):
            raise FileNotFoundError(
                f"dataset_path {root!r} does not exist (set DATASET_DIR or "
                "use --synthetic_data for benchmarking)")
        if getattr(self.args, "sets_are_pre_split", False):
            sets = {}
            for s in ("train", "val", "test"):
                sub = os.path.join(root, s)
                sets[s] = scan_class_folders(sub, self.args.indexes_of_folders_indicating_class)
        else:
            classes = scan_class_folders(root, self.args.indexes_of_folders_indicating_class)
            sets = split_classes(classes, self.args.train_val_test_split)
        try:
            with open(cache, "w") as f:
                json.dump(sets, f)
        except OSError:
            pass  # read-only dataset locations are fine, just skip the cache
        return sets

    # ------------------------------------------------------------------
    def _load_npz(self, npz_path: str) -> Dict[str, Dict[str, List[int]]]:
        """Compact preprocessed dataset (tools/make_omniglot_npz.py):
        images [num_classes, samples_per_class, H, W] uint8 + class names.
        Per-class 'file lists' become sample-index lists."""
        data = np.load(npz_path, allow_pickle=True)
        self._npz_images = data["images"]
        names = [str(n) for n in data["class_names"]]
        self._npz_class_index = {n: i for i, n in enumerate(names)}
        spc = self._npz_images.shape[1]
        classes = {n: list(range(spc)) for n in names}
        return split_classes(classes, self.args.train_val_test_split)

    def _npz_image(self, cname: str, index: int) -> np.ndarray:
        arr = self._npz_images[self._npz_class_index[cname], index]
        if arr.shape != (self.image_height, self.image_width):
            from PIL import Image
            img = Image.fromarray(arr, mode="L").resize(
                (self.image_width, self.image_height), Image.LANCZOS)
            arr = np.asarray(img)
        out = arr.astype(np.float32) / 255.0
        return out[:, :, None]

    def _load_image(self, path: str) -> np.ndarray:
        """Decode + resize one image -> float32 HWC in [0, 1]
        (reference: ``data.py:374-395``).  Corrupted files are tolerated
        with a zero image + warning (the reference detects and re-saves,
        ``data.py:280-300``; we cannot rewrite read-only datasets)."""
        from PIL import Image

        try:
            img = Image.open(path)
            img.load()
        except Exception:  # noqa: BLE001 - any decode failure
            import warnings
            warnings.warn(f"corrupted image {path!r}; substituting zeros")
            return np.zeros((self.image_height, self.image_width,
                             self.image_channels), dtype=np.float32)
        if self.image_channels == 1:
            img = img.convert("L")
        else:
            img = img.convert("RGB")
        img = img.resize((self.image_width, self.image_height), Image.LANCZOS)
        arr = np.asarray(img, dtype=np.float32) / 255.0
        if arr.ndim == 2:
            arr = arr[:, :, None]
        return arr

    def _preload(self) -> None:
        self._memory = {}
        paths = []
        keys = []
        for cname, files in self.datasets[self.current_set].items():
            for i, p in enumerate(files):
                paths.append(p)
                keys.append((cname, i))
        with concurrent.futures.ThreadPoolExecutor(
                max_workers=self.args.num_dataprovider_workers or 4) as ex:
            images = list(ex.map(self._load_image, paths))
        store: Dict[str, List[np.ndarray]] = {}
        for (cname, i), img in zip(keys, images):
            store.setdefault(cname, []).append(img)
        self._memory = store

    def _get_image(self, cname: str, index: int) -> np.ndarray:
        if self._npz_images is not None:
            return self._npz_image(cname, index)
        if self._memory is not None and cname in self._memory:
            return self._memory[cname][index]
        return self._load_image(self.datasets[self.current_set][cname][index])

    # ------------------------------------------------------------------
    def switch_set(self, set_name: str, current_iter: Optional[int] = None) -> None:
        """Reference: ``data.py:536-542`` — the train stream seed is rebased
        to ``init_seed + current_iter`` so training is resumable."""
        self.current_set = set_name
        if set_name == "train" and current_iter is not None:
            self.seed["train"] = self.init_seed["train"] + current_iter

    def get_set(self, set_name: str, seed: int, augment: bool = False):
        rng = np.random.RandomState(seed)
        classes = sorted(self.datasets[set_name].keys())
        chosen = rng.choice(len(classes), size=self.num_classes_per_set, replace=False)
        # per-class rot90 is the omniglot-style augmentation (reference
        # data.py:92-95); cifar datasets augment by crop/flip instead
        rotate = augment and "cifar" not in self.dataset_name.lower()
        k_per_class = rng.randint(0, 4, size=self.num_classes_per_set) if rotate \
            else np.zeros(self.num_classes_per_set, dtype=np.int64)
        s, t = self.num_samples_per_class, self.num_target_samples
        images = np.zeros((self.num_classes_per_set, s + t, self.image_height,
                           self.image_width, self.image_channels), dtype=np.float32)
        for ci, cidx in enumerate(chosen):
            cname = classes[int(cidx)]
            files = self.datasets[set_name][cname]
            sel = rng.choice(len(files), size=s + t, replace=len(files) < s + t)
            for si, fi in enumerate(sel):
                img = self._get_image(cname, int(fi))
                if k_per_class[ci]:
                    img = np.rot90(img, k=int(k_per_class[ci]), axes=(0, 1)).copy()
                images[ci, si] = img
        if self.image_channels == 3 and "imagenet" in self.dataset_name:
            images = (images - IMAGENET_MEAN) / IMAGENET_STD
        elif "cifar" in self.dataset_name:
            # reference transform set (data.py:81-90): train = RandomCrop
            # 32/pad4 + RandomHorizontalFlip + Normalize(classification_
            # mean/std); eval = Normalize only.  Crop/flip draws come from
            # the episode RNG so episodes stay a pure function of the seed.
            if augment:
                h, w = self.image_height, self.image_width
                for ci in range(images.shape[0]):
                    for si in range(images.shape[1]):
                        img = images[ci, si]
                        padded = np.zeros((h + 8, w + 8, img.shape[2]),
                                          dtype=img.dtype)
                        padded[4:4 + h, 4:4 + w] = img
                        oy, ox = rng.randint(0, 9), rng.randint(0, 9)
                        img = padded[oy:oy + h, ox:ox + w]
                        if rng.rand() < 0.5:
                            img = img[:, ::-1]
                        images[ci, si] = img
            mean = np.asarray(getattr(self.args, "classification_mean",
                                      [0.0, 0.0, 0.0]), dtype=np.float32)
            std = np.asarray(getattr(self.args, "classification_std",
                                     [1.0, 1.0, 1.0]), dtype=np.float32)
            images = (images - mean) / std
        # HWC -> CHW
        x = torch.from_numpy(images).permute(0, 1, 4, 2, 3).contiguous()
        labels = torch.arange(self.num_classes_per_set).view(-1, 1).expand(
            self.num_classes_per_set, s + t).contiguous()
        x_support, x_target = x[:, :s], x[:, s:]
        y_support, y_target = labels[:, :s], labels[:, s:]
        return x_support, x_target, y_support, y_target, seed

    def __len__(self) -> int:
        return int(self.args.total_epochs * self.args.total_iter_per_epoch *
                   self.args.batch_size)

    def __getitem__(self, idx: int):
        return self.get_set(self.current_set, seed=self.seed[self.current_set] + idx,
                            augment=self.augment_images)
