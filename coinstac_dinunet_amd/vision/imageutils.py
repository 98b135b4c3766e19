"""2D image utilities for vision computations.

Capability-parity with /root/reference/coinstac_dinunet/vision/imageutils.py:21-348
(Image loader with mask/ground-truth companions + enhancement, array PRF1A,
rescale, U-Net style patch chunking/merging with mirrored expansion,
connected-component cleanup, pixel neighborhood). Implemented on
PIL + numpy + scipy (no OpenCV dependency in this image); CLAHE is a
scipy-based adaptive histogram equalization.
"""
import math as _math

import numpy as _np

try:
    from PIL import Image as _PILImage
    _HAS_PIL = True
except ImportError:
    _HAS_PIL = False

try:
    from scipy import ndimage as _ndi
    _HAS_SCIPY = True
except ImportError:
    _HAS_SCIPY = False


class Image:
    """Image + optional mask/ground-truth companions."""

    def __init__(self, dir=None, file=None):
        self.dir = dir
        self.file = file
        self.array = None
        self.mask = None
        self.ground_truth = None
        self.extras = {}

    def load(self, dir=None, file=None):
        assert _HAS_PIL, 'PIL is required for Image.load'
        self.dir = dir or self.dir
        self.file = file or self.file
        self.array = _np.array(_PILImage.open(f'{self.dir}/{self.file}'))
        return self

    def load_mask(self, mask_dir, fget_mask=lambda f: f):
        assert _HAS_PIL, 'PIL is required for Image.load_mask'
        self.mask = _np.array(
            _PILImage.open(f'{mask_dir}/{fget_mask(self.file)}').convert('L'))
        return self

    def load_ground_truth(self, gt_dir, fget_ground_truth=lambda f: f):
        assert _HAS_PIL, 'PIL is required for Image.load_ground_truth'
        self.ground_truth = _np.array(
            _PILImage.open(f'{gt_dir}/{fget_ground_truth(self.file)}').convert('L'))
        return self

    def apply_mask(self):
        if self.mask is not None and self.array is not None:
            if self.array.ndim == 3:
                self.array[self.mask == 0] = 0
            else:
                self.array = self.array * (self.mask > 0)
        return self

    def apply_clahe(self, clip_limit=2.0, grid=(8, 8)):
        self.array = clahe_equalize(self.array, clip_limit=clip_limit,
                                    grid=grid)
        return self


def clahe_equalize(img, clip_limit=2.0, grid=(8, 8)):
    """Contrast-limited adaptive histogram equalization (single channel or
    per-channel), numpy implementation."""
    img = _np.asarray(img)
    if img.ndim == 3:
        out = img.copy()
        for c in range(img.shape[2]):
            out[..., c] = clahe_equalize(img[..., c], clip_limit, grid)
        return out
    img = img.astype(_np.uint8)
    h, w = img.shape
    gh, gw = grid
    th, tw = _math.ceil(h / gh), _math.ceil(w / gw)
    # per-tile clipped CDF lookup tables
    luts = _np.zeros((gh, gw, 256), dtype=_np.float32)
    for i in range(gh):
        for j in range(gw):
            tile = img[i * th:(i + 1) * th, j * tw:(j + 1) * tw]
            hist = _np.bincount(tile.reshape(-1), minlength=256).astype(_np.float64)
            if tile.size == 0:
                luts[i, j] = _np.arange(256)
                continue
            limit = max(1.0, clip_limit * tile.size / 256.0)
            excess = _np.clip(hist - limit, 0, None).sum()
            hist = _np.minimum(hist, limit) + excess / 256.0
            cdf = hist.cumsum()
            luts[i, j] = (cdf - cdf.min()) / max(cdf.max() - cdf.min(), 1) * 255.0
    # bilinear interpolation between tile LUTs
    yy, xx = _np.mgrid[0:h, 0:w]
    ty = (yy + 0.5) / th - 0.5
    tx = (xx + 0.5) / tw - 0.5
    y0 = _np.clip(_np.floor(ty).astype(int), 0, gh - 1)
    x0 = _np.clip(_np.floor(tx).astype(int), 0, gw - 1)
    y1 = _np.clip(y0 + 1, 0, gh - 1)
    x1 = _np.clip(x0 + 1, 0, gw - 1)
    wy = _np.clip(ty - y0, 0, 1)
    wx = _np.clip(tx - x0, 0, 1)
    v = img
    out = ((1 - wy) * (1 - wx) * luts[y0, x0, v] +
           (1 - wy) * wx * luts[y0, x1, v] +
           wy * (1 - wx) * luts[y1, x0, v] +
           wy * wx * luts[y1, x1, v])
    return out.astype(_np.uint8)


def rescale2d(arr, lo=0.0, hi=255.0):
    arr = arr.astype(_np.float64)
    mn, mx = arr.min(), arr.max()
    if mx - mn < 1e-12:
        return _np.full_like(arr, lo)
    return (arr - mn) / (mx - mn) * (hi - lo) + lo


def get_praf1(pred, true, eps=1e-5):
    """Precision/recall/accuracy/F1 on binary arrays."""
    pred = _np.asarray(pred).astype(bool)
    true = _np.asarray(true).astype(bool)
    tp = int((pred & true).sum())
    fp = int((pred & ~true).sum())
    fn = int((~pred & true).sum())
    tn = int((~pred & ~true).sum())
    p = tp / max(tp + fp, eps)
    r = tp / max(tp + fn, eps)
    a = (tp + tn) / max(tp + fp + fn + tn, eps)
    f1 = (2 * p * r) / max(p + r, eps)
    return {'precision': p, 'recall': r, 'accuracy': a, 'f1': f1}


def expand_and_mirror_patch(image, patch_region, pad):
    """Mirror-pad the patch window (U-Net style context tiling)."""
    (r0, r1), (c0, c1) = patch_region
    pr0, pr1, pc0, pc1 = pad
    padded = _np.pad(image, ((pr0, pr1), (pc0, pc1)), mode='reflect')
    return padded[r0:r1 + pr0 + pr1, c0:c1 + pc0 + pc1]


def get_chunk_indexes(img_shape, chunk_shape, offset=None):
    """Tile (row, col) windows covering img_shape with the given stride."""
    h, w = img_shape[:2]
    ch, cw = chunk_shape
    oh, ow = offset if offset else (ch, cw)
    rows = list(range(0, max(h - ch, 0) + 1, oh))
    cols = list(range(0, max(w - cw, 0) + 1, ow))
    if rows[-1] + ch < h:
        rows.append(h - ch)
    if cols[-1] + cw < w:
        cols.append(w - cw)
    return [[r, r + ch, c, c + cw] for r in rows for c in cols]


def merge_patches(patches, image_size, patch_size, offset=None):
    """Average-merge overlapping patches back to an image."""
    out = _np.zeros(image_size, dtype=_np.float64)
    count = _np.zeros(image_size, dtype=_np.float64)
    for patch, (r0, r1, c0, c1) in zip(
            patches, get_chunk_indexes(image_size, patch_size, offset)):
        out[r0:r1, c0:c1] += patch
        count[r0:r1, c0:c1] += 1
    return out / _np.clip(count, 1, None)


def largest_cc(binary_arr):
    """Keep only the largest connected component."""
    assert _HAS_SCIPY, 'scipy is required for largest_cc'
    labels, n = _ndi.label(_np.asarray(binary_arr).astype(bool))
    if n == 0:
        return _np.zeros_like(labels, dtype=bool)
    sizes = _ndi.sum(_np.ones_like(labels), labels, index=range(1, n + 1))
    return labels == (int(_np.argmax(sizes)) + 1)


def remove_small_cc(binary_arr, min_size=64):
    """Drop connected components smaller than min_size pixels."""
    assert _HAS_SCIPY, 'scipy is required for remove_small_cc'
    arr = _np.asarray(binary_arr).astype(bool)
    labels, n = _ndi.label(arr)
    if n == 0:
        return arr
    sizes = _np.bincount(labels.reshape(-1))
    keep = sizes >= min_size
    keep[0] = False
    return keep[labels]


def get_pix_neigh(i, j, eight=True):
    """Pixel neighborhood coordinates (4- or 8-connectivity)."""
    n4 = [(i - 1, j), (i + 1, j), (i, j - 1), (i, j + 1)]
    if not eight:
        return n4
    return n4 + [(i - 1, j - 1), (i - 1, j + 1), (i + 1, j - 1), (i + 1, j + 1)]
