"""Vision layer: plotter output, image utilities."""
import numpy as np
import pytest

from coinstac_dinunet_amd.vision import imageutils, plotter


def test_plot_progress_writes_png(tmp_path):
    cache = {'log_header': 'Loss|Accuracy,F1',
             'train_log': [[1.0, 0.5, 0.4], [0.8, 0.6, 0.5],
                           [0.6, 0.7, 0.6], [0.5, 0.75, 0.7]]}
    plotter.plot_progress(cache, str(tmp_path), plot_keys=['train_log'])
    assert (tmp_path / 'train_log.png').exists()


def test_get_praf1():
    pred = np.array([[1, 0], [1, 1]])
    true = np.array([[1, 0], [0, 1]])
    r = imageutils.get_praf1(pred, true)
    assert r['precision'] == pytest.approx(2 / 3, abs=1e-4)
    assert r['recall'] == pytest.approx(1.0, abs=1e-4)
    assert r['accuracy'] == pytest.approx(0.75, abs=1e-4)


def test_rescale2d():
    arr = np.array([[0.0, 5.0], [10.0, 2.5]])
    out = imageutils.rescale2d(arr, 0, 255)
    assert out.min() == 0 and out.max() == 255


def test_chunk_and_merge_roundtrip():
    img = np.arange(64, dtype=np.float64).reshape(8, 8)
    idx = imageutils.get_chunk_indexes((8, 8), (4, 4))
    patches = [img[r0:r1, c0:c1] for r0, r1, c0, c1 in idx]
    merged = imageutils.merge_patches(patches, (8, 8), (4, 4))
    np.testing.assert_allclose(merged, img)


def test_largest_and_small_cc():
    arr = np.zeros((10, 10), dtype=bool)
    arr[0:2, 0:2] = True   # size 4
    arr[5:9, 5:9] = True   # size 16
    big = imageutils.largest_cc(arr)
    assert big.sum() == 16
    cleaned = imageutils.remove_small_cc(arr, min_size=5)
    assert cleaned.sum() == 16


def test_clahe_shapes():
    img = (np.random.RandomState(0).rand(32, 32) * 255).astype(np.uint8)
    out = imageutils.clahe_equalize(img)
    assert out.shape == img.shape and out.dtype == np.uint8


def test_pix_neigh():
    assert len(imageutils.get_pix_neigh(3, 3)) == 8
    assert len(imageutils.get_pix_neigh(3, 3, eight=False)) == 4


def test_image_loader_roundtrip(tmp_path):
    from PIL import Image as PILImage
    from coinstac_dinunet_amd.vision.imageutils import Image
    arr = (np.random.RandomState(1).rand(16, 16, 3) * 255).astype(np.uint8)
    PILImage.fromarray(arr).save(tmp_path / 'img.png')
    mask = (np.random.RandomState(2).rand(16, 16) > 0.5).astype(np.uint8) * 255
    PILImage.fromarray(mask).save(tmp_path / 'img_mask.png')
    im = Image(dir=str(tmp_path), file='img.png').load()
    assert im.array.shape == (16, 16, 3)
    im.load_mask(str(tmp_path), fget_mask=lambda f: f.replace('.png', '_mask.png'))
    im.apply_mask()
    assert (im.array[mask == 0] == 0).all()
