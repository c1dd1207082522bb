"""ImageSaver — dump misclassified/worst samples as images per epoch.

Parity target: reference ``znicz/image_saver.py`` (``ImageSaver``):
writes the minibatch samples the evaluator got wrong into
``out_dirs[class]`` with the (label, predicted, index) encoded in the
filename.  Uses PIL when the sample looks like an image, ``.npy``
otherwise.
"""

import os

import numpy

from veles.config import root
from veles.units import Unit


class ImageSaver(Unit):
    def __init__(self, workflow, **kwargs):
        super(ImageSaver, self).__init__(workflow, **kwargs)
        base = kwargs.get("out_dirs", None)
        if base is None:
            d = os.path.join(root.common.dirs.get("base", "."),
                             "image_saver")
            base = [os.path.join(d, s) for s in
                    ("test", "validation", "train")]
        self.out_dirs = list(base)
        self.limit = kwargs.get("limit", 64)
        # linked attrs
        self.input = None               # minibatch_data
        self.labels = None              # minibatch_labels
        self.max_idx = None             # argmax of the softmax output
        self.indices = None             # minibatch_indices
        self.minibatch_class = 0
        self.minibatch_size = 0
        self.this_save_time = 0
        self._saved = 0

    def initialize(self, **kwargs):
        super(ImageSaver, self).initialize(**kwargs)
        for d in self.out_dirs:
            os.makedirs(d, exist_ok=True)

    def reset_counter(self):
        self._saved = 0

    def run(self):
        if self._saved >= self.limit:
            return
        for arr in (self.input, self.labels, self.max_idx):
            if arr is None:
                return
            arr.map_read()
        if self.indices is not None:
            self.indices.map_read()
        n = int(self.minibatch_size)
        labels = self.labels.mem
        preds = self.max_idx.mem
        data = self.input.mem
        outd = self.out_dirs[min(int(self.minibatch_class),
                                 len(self.out_dirs) - 1)]
        for i in range(n):
            if self._saved >= self.limit:
                break
            if int(labels[i]) == int(preds[i]):
                continue
            idx = int(self.indices.mem[i]) if self.indices is not None \
                else i
            name = "%d_as_%d_idx%d" % (int(labels[i]), int(preds[i]),
                                       idx)
            self._save_sample(data[i], os.path.join(outd, name))
            self._saved += 1

    def _save_sample(self, sample, path_base):
        img = numpy.asarray(sample)
        if img.ndim == 3 and img.shape[-1] in (1, 3, 4, 8):
            img2 = img[..., :3] if img.shape[-1] >= 3 else img[..., 0]
        elif img.ndim == 2:
            img2 = img
        else:
            numpy.save(path_base + ".npy", img)
            return
        try:
            from PIL import Image
            lo, hi = float(img2.min()), float(img2.max())
            scaled = ((img2 - lo) / (hi - lo + 1e-12) * 255).astype(
                numpy.uint8)
            Image.fromarray(scaled).save(path_base + ".png")
        except Exception:
            numpy.save(path_base + ".npy", img)
