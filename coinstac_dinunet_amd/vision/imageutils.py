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


def rescale2d(arr, lo=0.0, hi=1.0):
    """Min-max rescale to [lo, hi] (reference default: unit range,
    imageutils.py:154-157)."""
    arr = arr.astype(_np.float64)
    mn, mx = arr.min(), arr.max()
    if mx - mn < 1e-12:
        return _np.full_like(arr, lo)
    return (arr - mn) / (mx - mn) * (hi - lo) + lo


def rescale3d(arrays):
    """Per-slice unit rescale (reference imageutils.py:160-161)."""
    return [rescale2d(a) for a in arrays]


def get_signed_diff_int8(image_arr1, image_arr2):
    """Signed pixel difference rescaled to uint8 for visual diffing
    (reference imageutils.py:164-168)."""
    signed = _np.asarray(image_arr1 - image_arr2, dtype=_np.int8)
    shifted = _np.asarray(signed - signed.min(), dtype=_np.uint8)
    return _np.asarray(rescale2d(shifted) * 255, dtype=_np.uint8)


def whiten_image2d(img_arr2d):
    """Zero-mean/unit-std whiten, then stretch back to uint8
    (reference imageutils.py:171-174)."""
    z = (img_arr2d - img_arr2d.mean()) / max(float(img_arr2d.std()), 1e-12)
    return _np.asarray(rescale2d(z) * 255, dtype=_np.uint8)


def _binarize_255(arr):
    a = _np.asarray(arr).copy()
    a[a == 255] = 1
    return a.astype(_np.int64)


def get_rgb_scores(arr_2d, truth):
    """RGB overlay of prediction vs ground truth: white=TP, green=FP,
    red=FN, black=TN (reference imageutils.py:88-107)."""
    xy = _binarize_255(arr_2d) + 2 * _binarize_255(truth)
    rgb = _np.zeros([xy.shape[0], xy.shape[1], 3], dtype=_np.uint8)
    rgb[xy == 3] = [255, 255, 255]
    rgb[xy == 1] = [0, 255, 0]
    rgb[xy == 2] = [255, 0, 0]
    return rgb


def get_praf1(arr_2d, truth):
    """Precision/recall/accuracy/F1 between binary (0/1 or 0/255) arrays,
    rounded to 5 places (reference imageutils.py:110-151)."""
    xy = _binarize_255(arr_2d) + 2 * _binarize_255(truth)
    tp = int((xy == 3).sum())
    fp = int((xy == 1).sum())
    fn = int((xy == 2).sum())
    tn = int((xy == 0).sum())
    p = tp / (tp + fp) if tp + fp else 0
    r = tp / (tp + fn) if tp + fn else 0
    a = (tp + tn) / (tp + fp + fn + tn) if tp + fp + fn + tn else 0
    f1 = 2 * p * r / (p + r) if p + r else 0
    return {'Precision': round(p, 5), 'Recall': round(r, 5),
            'Accuracy': round(a, 5), 'F1': round(f1, 5)}


def map_img_to_img2d(map_to, img):
    """Burn a binary overlay into an image (gray promoted to RGB; overlay
    pixels turn red — reference imageutils.py:290-301)."""
    arr = _np.asarray(map_to).copy()
    if arr.ndim == 2:
        rgb = _np.stack([arr, arr, arr], -1).astype(_np.uint8)
    else:
        rgb = arr.astype(_np.uint8)
    on = _np.asarray(img) == 255
    rgb[..., 0][on] = 255
    rgb[..., 1][on] = 0
    rgb[..., 2][on] = 0
    return rgb


def expand_and_mirror_patch(full_img_shape, orig_patch_indices, expand_by):
    """Expanded window around a patch for U-Net-style context tiling:
    returns the clamped window indices plus the reflect-pad spec covering
    whatever margin fell outside the image (reference
    imageutils.py:253-279). Apply via
    np.pad(img[a:b, c:d], pad_spec, mode='reflect')."""
    half_i, half_j = int(expand_by[0] / 2), int(expand_by[1] / 2)
    p, q, r, s = orig_patch_indices
    a, b, c, d = p - half_i, q + half_i, r - half_j, s + half_j
    pad = [0, 0, 0, 0]
    if a < 0:
        pad[0], a = half_i - p, 0
    if b > full_img_shape[0]:
        pad[1], b = b - full_img_shape[0], full_img_shape[0]
    if c < 0:
        pad[2], c = half_j - r, 0
    if d > full_img_shape[1]:
        pad[3], d = d - full_img_shape[1], full_img_shape[1]
    return a, b, c, d, [(pad[0], pad[1]), (pad[2], pad[3])]


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


def get_pix_neigh(i, j, eight=False):
    """Pixel neighborhood coordinates (4- or 8-connectivity; reference
    imageutils.py:328-348 defaults to 4)."""
    n4 = [(i - 1, j), (i + 1, j), (i, j - 1), (i, j + 1)]
    if not eight:
        return n4
    return n4 + [(i - 1, j - 1), (i - 1, j + 1), (i + 1, j - 1), (i + 1, j + 1)]


def get_chunk_indices_by_index(img_shape, chunk_shape, indices):
    """Windows of chunk_shape centered on the given (row, col) points,
    clamped inside the image (reference imageutils.py:211-226)."""
    x, y = chunk_shape
    out = []
    w, h = img_shape[:2]
    for (c1, c2) in indices:
        p, q = c1 - x // 2, c1 + x // 2
        r, s = c2 - y // 2, c2 + y // 2
        if p < 0:
            p, q = 0, x
        if q > w:
            p, q = w - x, w
        if r < 0:
            r, s = 0, y
        if s > h:
            r, s = h - y, h
        out.append([int(p), int(q), int(r), int(s)])
    return out


def remove_connected_comp(segmented_img, connected_comp_diam_limit=20):
    """Zero out connected components whose bounding-box diagonal is below
    the diameter limit (reference imageutils.py:304-325; vectorized via
    per-label bounding boxes instead of a per-pixel loop)."""
    assert _HAS_SCIPY, 'scipy is required for remove_connected_comp'
    img = _np.asarray(segmented_img).copy()
    labeled, n = _ndi.label(img, _np.ones((3, 3), dtype=int))
    if n == 0:
        return img
    for sl, lab in zip(_ndi.find_objects(labeled), range(1, n + 1)):
        if sl is None:
            continue
        dy = sl[0].stop - sl[0].start - 1
        dx = sl[1].stop - sl[1].start - 1
        if _math.sqrt(dy * dy + dx * dx) < connected_comp_diam_limit:
            img[labeled == lab] = 0
    return img
