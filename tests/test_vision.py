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
    assert r['Precision'] == pytest.approx(2 / 3, abs=1e-4)
    assert r['Recall'] == pytest.approx(1.0, abs=1e-4)
    assert r['Accuracy'] == pytest.approx(0.75, abs=1e-4)
    # 0/255 binary images behave identically
    r255 = imageutils.get_praf1(pred * 255, true * 255)
    assert r255 == r


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
    assert len(imageutils.get_pix_neigh(3, 3, eight=True)) == 8
    assert len(imageutils.get_pix_neigh(3, 3)) == 4  # reference default


def test_rgb_scores_color_coding():
    pred = np.array([[1, 1], [0, 0]])
    true = np.array([[1, 0], [1, 0]])
    rgb = imageutils.get_rgb_scores(pred, true)
    assert (rgb[0, 0] == [255, 255, 255]).all()  # TP white
    assert (rgb[0, 1] == [0, 255, 0]).all()      # FP green
    assert (rgb[1, 0] == [255, 0, 0]).all()      # FN red
    assert (rgb[1, 1] == [0, 0, 0]).all()        # TN black


def test_whiten_and_signed_diff():
    img = (np.random.RandomState(3).rand(8, 8) * 200).astype(np.uint8)
    w = imageutils.whiten_image2d(img)
    assert w.dtype == np.uint8 and w.min() == 0 and w.max() == 255
    d = imageutils.get_signed_diff_int8(img, img)
    assert d.shape == img.shape
    r3 = imageutils.rescale3d([img, img * 0 + 7])
    assert len(r3) == 2 and r3[0].max() == pytest.approx(1.0)


def test_chunk_indices_by_index_clamped():
    ix = imageutils.get_chunk_indices_by_index((16, 16), (8, 8),
                                               [(0, 0), (8, 8), (15, 15)])
    for p, q, r, s in ix:
        assert 0 <= p < q <= 16 and q - p == 8
        assert 0 <= r < s <= 16 and s - r == 8


def test_expand_and_mirror_patch():
    # interior patch: no padding needed
    a, b, c, d, pad = imageutils.expand_and_mirror_patch((32, 32),
                                                         (8, 16, 8, 16), (8, 8))
    assert (a, b, c, d) == (4, 20, 4, 20) and pad == [(0, 0), (0, 0)]
    # corner patch: clipped margin comes back as reflect padding
    a, b, c, d, pad = imageutils.expand_and_mirror_patch((32, 32),
                                                         (0, 8, 0, 8), (8, 8))
    assert (a, c) == (0, 0) and pad == [(4, 0), (4, 0)]
    window = np.pad(np.zeros((b - a, d - c)), pad, mode='reflect')
    assert window.shape == (16, 16)


def test_map_img_and_remove_diam():
    base = np.full((10, 10), 100, dtype=np.uint8)
    overlay = np.zeros((10, 10), dtype=np.uint8)
    overlay[2, 2] = 255
    rgb = imageutils.map_img_to_img2d(base, overlay)
    assert (rgb[2, 2] == [255, 0, 0]).all()
    assert (rgb[0, 0] == [100, 100, 100]).all()
    seg = np.zeros((20, 20), dtype=np.uint8)
    seg[1, 1] = 1              # diameter ~0 -> removed
    seg[5:15, 5] = 1           # tall component, diagonal ~9 -> kept at limit 5
    out = imageutils.remove_connected_comp(seg, connected_comp_diam_limit=5)
    assert out[1, 1] == 0 and out[10, 5] == 1


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
