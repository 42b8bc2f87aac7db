"""Self-contained ImageFolder + ImageNet transforms (PIL + torch).

The reference takes torchvision's ImageFolder/transforms
(run_vit_training.py:40-55: RandomResizedCrop+flip for train,
Resize(256/224·S)+CenterCrop for val, bicubic, ImageNet mean/std).
torchvision is not installed in this image, so the same semantics are
implemented directly on PIL + torch — same sampling algorithm for
RandomResizedCrop (10 area/log-ratio attempts, center fallback), same
short-side Resize rounding, bicubic everywhere, float [0,1] CHW then
mean/std normalize.
"""

import math
import os

import torch

IMG_EXTENSIONS = (".jpg", ".jpeg", ".png", ".bmp", ".webp", ".ppm", ".tif",
                  ".tiff")


def _pil():
    from PIL import Image

    return Image


class ImageFolder(torch.utils.data.Dataset):
    """`root/<class_name>/<image>` layout, classes sorted by name
    (torchvision-compatible class indexing)."""

    def __init__(self, root, transform=None):
        self.root = root
        self.transform = transform
        classes = sorted(
            d for d in os.listdir(root)
            if os.path.isdir(os.path.join(root, d))
        )
        if not classes:
            raise FileNotFoundError(f"no class directories under {root}")
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = []
        for c in classes:
            cdir = os.path.join(root, c)
            for dirpath, _, files in sorted(os.walk(cdir)):
                for fname in sorted(files):
                    if fname.lower().endswith(IMG_EXTENSIONS):
                        self.samples.append(
                            (os.path.join(dirpath, fname),
                             self.class_to_idx[c])
                        )
        if not self.samples:
            raise FileNotFoundError(f"no images found under {root}")

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        path, target = self.samples[idx]
        img = _pil().open(path).convert("RGB")
        if self.transform is not None:
            img = self.transform(img)
        return img, target

    def __repr__(self):
        return (f"ImageFolder(root={self.root!r}, images={len(self)}, "
                f"classes={len(self.class_to_idx)})")


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class RandomResizedCrop:
    """torchvision's sampling algorithm: up to 10 attempts drawing a
    target area in `scale`·area and a log-uniform aspect ratio in
    `ratio`; fallback = largest in-ratio center crop."""

    def __init__(self, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3)):
        self.size = size
        self.scale = scale
        self.ratio = ratio

    def _params(self, w, h):
        area = w * h
        log_ratio = (math.log(self.ratio[0]), math.log(self.ratio[1]))
        for _ in range(10):
            target = area * (
                self.scale[0]
                + (self.scale[1] - self.scale[0]) * torch.rand(1).item()
            )
            aspect = math.exp(
                log_ratio[0]
                + (log_ratio[1] - log_ratio[0]) * torch.rand(1).item()
            )
            cw = int(round(math.sqrt(target * aspect)))
            ch = int(round(math.sqrt(target / aspect)))
            if 0 < cw <= w and 0 < ch <= h:
                left = torch.randint(0, w - cw + 1, (1,)).item()
                top = torch.randint(0, h - ch + 1, (1,)).item()
                return left, top, cw, ch
        # center fallback, clamped into the ratio range
        in_ratio = w / h
        if in_ratio < self.ratio[0]:
            cw, ch = w, int(round(w / self.ratio[0]))
        elif in_ratio > self.ratio[1]:
            cw, ch = int(round(h * self.ratio[1])), h
        else:
            cw, ch = w, h
        return (w - cw) // 2, (h - ch) // 2, cw, ch

    def __call__(self, img):
        w, h = img.size
        left, top, cw, ch = self._params(w, h)
        img = img.crop((left, top, left + cw, top + ch))
        return img.resize((self.size, self.size), _pil().Resampling.BICUBIC)


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if torch.rand(1).item() < self.p:
            return img.transpose(_pil().Transpose.FLIP_LEFT_RIGHT)
        return img


class Resize:
    """Short side -> `size`, aspect preserved (torchvision int-size
    semantics), bicubic."""

    def __init__(self, size):
        self.size = size

    def __call__(self, img):
        w, h = img.size
        if w <= h:
            nw, nh = self.size, max(1, int(round(self.size * h / w)))
        else:
            nw, nh = max(1, int(round(self.size * w / h))), self.size
        return img.resize((nw, nh), _pil().Resampling.BICUBIC)


class CenterCrop:
    def __init__(self, size):
        self.size = size

    def __call__(self, img):
        w, h = img.size
        left = (w - self.size) // 2
        top = (h - self.size) // 2
        return img.crop((left, top, left + self.size, top + self.size))


class ToTensor:
    """PIL RGB -> float32 CHW in [0, 1]."""

    def __call__(self, img):
        t = torch.frombuffer(
            bytearray(img.tobytes()), dtype=torch.uint8
        ).clone()
        t = t.view(img.size[1], img.size[0], 3).permute(2, 0, 1)
        return t.float().div_(255.0)


class Normalize:
    def __init__(self, mean, std):
        self.mean = torch.tensor(mean).view(3, 1, 1)
        self.std = torch.tensor(std).view(3, 1, 1)

    def __call__(self, t):
        return (t - self.mean) / self.std
