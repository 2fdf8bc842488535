"""MultibatchData-style list-file dataset.

The reference's data layer reads `source` (a text file of
"<relative path> <label>" lines) under `root_folder`
(usage/def.prototxt:18-20).  This offline image has no JPEG decoder
(no PIL/cv2/torchvision), so supported payload formats are:
  .npy          — numpy array HxWx3 (uint8 or float) or 3xHxW float
  .pt           — torch tensor 3xHxW
  .ppm          — binary P6 portable pixmap (decoded here)
Resize-to-(new_height,new_width) uses torch interpolation.  The P x K
batch structure comes from PKBatchSampler, augmentation from
DataTransformer — this class only loads and normalizes shapes.
"""

from __future__ import annotations

import os
from typing import List, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset


def _load_ppm(path: str) -> np.ndarray:
    with open(path, "rb") as fh:
        data = fh.read()
    if not data.startswith(b"P6"):
        raise ValueError(f"{path}: only binary P6 PPM supported")
    # header: P6 <w> <h> <maxval>\n, with comment lines allowed
    parts: List[bytes] = []
    pos = 2
    while len(parts) < 3:
        while pos < len(data) and data[pos : pos + 1].isspace():
            pos += 1
        if data[pos : pos + 1] == b"#":
            while data[pos : pos + 1] != b"\n":
                pos += 1
            continue
        start = pos
        while pos < len(data) and not data[pos : pos + 1].isspace():
            pos += 1
        parts.append(data[start:pos])
    pos += 1  # the single whitespace after maxval
    w, h, maxval = int(parts[0]), int(parts[1]), int(parts[2])
    arr = np.frombuffer(data, dtype=np.uint8, count=w * h * 3, offset=pos)
    return arr.reshape(h, w, 3)


def _to_chw_float(x) -> torch.Tensor:
    if isinstance(x, np.ndarray):
        t = torch.from_numpy(np.ascontiguousarray(x))
    else:
        t = x
    t = t.float()
    if t.dim() == 3 and t.shape[-1] == 3 and t.shape[0] != 3:
        t = t.permute(2, 0, 1)  # HWC -> CHW
    if t.dim() != 3:
        raise ValueError(f"expected 3D image tensor, got {tuple(t.shape)}")
    return t


class FolderListDataset(Dataset):
    def __init__(self, root_folder: str, source: str,
                 new_height: int = 224, new_width: int = 224):
        self.root = root_folder
        self.size = (new_height, new_width)
        self.items: List[Tuple[str, int]] = []
        with open(source) as fh:
            for line in fh:
                line = line.strip()
                if not line:
                    continue
                path, lab = line.rsplit(None, 1)
                self.items.append((path, int(float(lab))))
        self.labels = [lab for _, lab in self.items]

    def __len__(self):
        return len(self.items)

    def __getitem__(self, idx):
        rel, lab = self.items[idx]
        path = os.path.join(self.root, rel)
        ext = os.path.splitext(path)[1].lower()
        if ext == ".npy":
            img = _to_chw_float(np.load(path))
        elif ext == ".pt":
            img = _to_chw_float(torch.load(path, weights_only=True))
        elif ext == ".ppm":
            img = _to_chw_float(_load_ppm(path))
        else:
            raise ValueError(f"unsupported image format {ext!r} (npy/pt/ppm)")
        if img.shape[1:] != self.size:
            img = torch.nn.functional.interpolate(
                img.unsqueeze(0), size=self.size, mode="bilinear",
                align_corners=False).squeeze(0)
        return img, lab
