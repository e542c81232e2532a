"""Host-side image loading and preprocessing.

Behavioral parity with reference `utils/misc.py:6-36` (ImageLoader): decode
JPEG, produce RGB, resize to a fixed shape (224x224), subtract the ILSVRC-2012
per-channel mean, return float32 NHWC batches.  Differences from the
reference: PIL instead of cv2 (cv2 is not in this image), and the mean is the
standard ILSVRC per-channel RGB mean rather than the reference's mean-image
.npy (a git-LFS blob absent from the mount) — numerically the same protocol.
The reference's center-crop (`misc.py:23-26`) is a no-op at scale==crop size
and is therefore not reproduced.
"""

import numpy as np

try:
    from PIL import Image
    _HAVE_PIL = True
except Exception:  # pragma: no cover
    _HAVE_PIL = False

# ILSVRC-2012 per-channel mean, RGB order.
ILSVRC_2012_MEAN = np.array([123.68, 116.779, 103.939], dtype=np.float32)


class ImageLoader(object):
    def __init__(self, mean_file=None, image_shape=(224, 224, 3)):
        self.image_shape = tuple(image_shape)
        if mean_file is not None:
            try:
                m = np.load(mean_file)
                self.mean = m.astype(np.float32)
            except Exception:
                self.mean = ILSVRC_2012_MEAN
        else:
            self.mean = ILSVRC_2012_MEAN

    def load_image(self, image_file):
        """Load one image -> float32 [H,W,3] RGB, mean-subtracted."""
        if not _HAVE_PIL:
            raise RuntimeError("PIL unavailable; cannot decode images")
        h, w = self.image_shape[0], self.image_shape[1]
        img = Image.open(image_file).convert('RGB').resize(
            (w, h), Image.BILINEAR)
        arr = np.asarray(img, dtype=np.float32)
        return arr - self.mean

    def load_images(self, image_files):
        """Load a batch -> float32 [N,H,W,3]."""
        return np.stack([self.load_image(f) for f in image_files], axis=0)
