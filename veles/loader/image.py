"""Image loading & augmentation.

Parity target: reference ``veles/loader/image.py`` (``ImageLoader``:
scale / crop / mirror / color-space options), ``file_image.py``
(``FileImageLoader``: directory globs → classes) and
``fullbatch_image.py`` (``FullBatchImageLoader``).

MI355X note: channels are padded to a multiple of 8 (``pad_channels``)
so the NHWC HIP conv path's 16-B DMA constraint holds for RGB input.
"""

import glob
import os

import numpy

from veles.loader.base import Loader
from veles.loader.fullbatch import FullBatchLoader
from veles.loader.fullbatch import FullBatchLoaderMSE as _FBMSE


def pad_channels_to8(img):
    """[H, W, C] -> [H, W, ceil8(C)] zero-padded."""
    h, w, c = img.shape
    c8 = (c + 7) & ~7
    if c8 == c:
        return img
    out = numpy.zeros((h, w, c8), img.dtype)
    out[:, :, :c] = img
    return out


class ImageLoader(Loader):
    """Base: decoding + augmentation pipeline for one image."""

    def __init__(self, workflow, **kwargs):
        super(ImageLoader, self).__init__(workflow, **kwargs)
        self.scale = kwargs.get("scale", None)        # (w, h) or float
        self.crop = kwargs.get("crop", None)          # (w, h) crop size
        #: "center" (deterministic) or "random" (augmentation; uses the
        #: loader's seeded rng so runs stay reproducible)
        self.crop_mode = kwargs.get("crop_mode", "center")
        self.mirror = kwargs.get("mirror", False)     # random h-flip
        self.grayscale = kwargs.get("grayscale", False)
        self.normalize_range = kwargs.get("normalize_range", (-1.0, 1.0))
        self.pad_channels = kwargs.get("pad_channels", True)

    # -- single-image pipeline ----------------------------------------------
    def decode_image(self, path):
        from PIL import Image
        img = Image.open(path)
        img = img.convert("L" if self.grayscale else "RGB")
        return img

    def transform_image(self, img, rng=None):
        from PIL import Image
        if self.scale is not None:
            if isinstance(self.scale, (int, float)):
                w, h = img.size
                size = (int(w * self.scale), int(h * self.scale))
            else:
                size = tuple(self.scale)
            img = img.resize(size, Image.BILINEAR)
        if self.crop is not None:
            cw, ch = self.crop
            w, h = img.size
            if self.crop_mode == "random" and rng is not None:
                x0 = rng.randint(0, max(1, w - cw + 1))
                y0 = rng.randint(0, max(1, h - ch + 1))
            else:
                x0, y0 = (w - cw) // 2, (h - ch) // 2
            img = img.crop((x0, y0, x0 + cw, y0 + ch))
        arr = numpy.asarray(img, numpy.float32)
        if arr.ndim == 2:
            arr = arr[:, :, None]
        if self.mirror and rng is not None and rng.randint(0, 2):
            arr = arr[:, ::-1, :]
        lo, hi = self.normalize_range
        arr = arr / 255.0 * (hi - lo) + lo
        if self.pad_channels:
            arr = pad_channels_to8(arr)
        return numpy.ascontiguousarray(arr)

    def load_image(self, path, rng=None):
        return self.transform_image(self.decode_image(path), rng)


class FileImageLoader(ImageLoader, FullBatchLoader):
    """Globs per-class directories, loads every image up front (the
    datasets the reference samples use all fit in memory; 288 GB HBM
    holds them device-resident afterwards)."""

    def __init__(self, workflow, **kwargs):
        super(FileImageLoader, self).__init__(workflow, **kwargs)
        # each: list of glob patterns, e.g. ["/data/train/*/*.png"]
        self.test_paths = kwargs.get("test_paths", [])
        self.validation_paths = kwargs.get("validation_paths", [])
        self.train_paths = kwargs.get("train_paths", [])
        self.label_from = kwargs.get("label_from", "dirname")
        self._label_map = {}

    def get_label(self, path):
        if callable(self.label_from):
            return self.label_from(path)
        key = os.path.basename(os.path.dirname(path))
        if key not in self._label_map:
            self._label_map[key] = len(self._label_map)
        return self._label_map[key]

    def _expand(self, patterns):
        files = []
        for pat in patterns:
            files.extend(sorted(glob.glob(pat)))
        return files

    def load_data(self):
        sets = [self._expand(p) for p in (self.test_paths,
                                          self.validation_paths,
                                          self.train_paths)]
        total = sum(len(s) for s in sets)
        if total == 0:
            raise ValueError("FileImageLoader found no files")
        first = self.load_image(sets[-1][0] if sets[-1] else
                                [f for s in sets for f in s][0])
        data = numpy.zeros((total,) + first.shape, numpy.float32)
        labels = numpy.zeros(total, numpy.int32)
        i = 0
        rng = numpy.random.RandomState(1)
        for s in sets:
            for path in s:
                data[i] = self.load_image(path, rng)
                labels[i] = self.get_label(path)
                i += 1
        self.original_data.mem = data
        self.original_labels.mem = labels
        self.class_lengths = [len(s) for s in sets]


FullBatchImageLoader = FileImageLoader  # reference alias


class FileImageLoaderMSE(FileImageLoader, _FBMSE):
    """Input-image / target-image dataset pairs for MSE training
    (reference ``fullbatch_image.py`` + ``image_mse.py``: denoising /
    reconstruction autoencoders train against a TARGET image per
    sample, not a class label).

    ``target_paths`` mirrors ``*_paths``: one glob list whose files
    pair with the input files by basename (the reference paired by
    file order within matching directory trees; basename pairing is
    the robust version of that contract).  Targets go through the same
    scale/crop pipeline (deterministic: no mirror/random-crop skew
    between input and target — augmentation is applied to neither).
    """

    def __init__(self, workflow, **kwargs):
        super(FileImageLoaderMSE, self).__init__(workflow, **kwargs)
        self.target_paths = kwargs.get("target_paths", [])

    def load_data(self):
        # inputs first (fills original_data + class_lengths)
        saved_mirror, self.mirror = self.mirror, False
        saved_mode, self.crop_mode = self.crop_mode, "center"
        try:
            super(FileImageLoaderMSE, self).load_data()
            tfiles = self._expand(self.target_paths)
            if not tfiles:
                raise ValueError("FileImageLoaderMSE: no target files")
            by_base = {os.path.basename(p): p for p in tfiles}
            sets = [self._expand(p) for p in (self.test_paths,
                                              self.validation_paths,
                                              self.train_paths)]
            order = [p for s in sets for p in s]
            first = self.load_image(tfiles[0])
            targets = numpy.zeros((len(order),) + first.shape,
                                  numpy.float32)
            for i, path in enumerate(order):
                base = os.path.basename(path)
                if base not in by_base:
                    raise ValueError(
                        "no target image pairs with input %r" % path)
                targets[i] = self.load_image(by_base[base])
            self.original_targets.mem = targets
        finally:
            self.mirror = saved_mirror
            self.crop_mode = saved_mode
    # minibatch_targets machinery (create/initialize/fill) comes from
    # FullBatchLoaderMSE via the MRO
