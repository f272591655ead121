"""Golden tests: CPU executor ops vs the original PIL implementations.

The reference applies these ops via PIL (reference augmentations.py:13-182);
our CPU executor re-implements the pixel math and the HIP kernels are tested
against the CPU executor, so PIL-exactness here anchors the whole chain.
"""
import numpy as np
import pytest

import PIL
import PIL.Image
import PIL.ImageDraw
import PIL.ImageEnhance
import PIL.ImageOps

from fast_autoaugment_amd.aug import cpu_exec
from fast_autoaugment_amd.aug.ops import _rotate_matrix


RNG = np.random.default_rng(1234)


def rand_img(h=32, w=32):
    return RNG.integers(0, 256, size=(h, w, 3), dtype=np.uint8)


def assert_affine_close(got, ref, m, h=32, w=32):
    """Strict equality except at pixels whose sampled coordinate sits on an
    integer boundary: PIL's C loop computes coords by repeated addition
    (accumulating fp error), we compute directly; on exact boundaries the
    nearest-neighbor pick can flip. Those pixels must still be rare."""
    a, b, c, d, e, f = m
    ys, xs = np.mgrid[0:h, 0:w]
    xin = a * (xs + 0.5) + b * (ys + 0.5) + c
    yin = d * (xs + 0.5) + e * (ys + 0.5) + f
    on_bx = np.abs(xin - np.round(xin)) < 1e-9
    on_by = np.abs(yin - np.round(yin)) < 1e-9
    boundary = (on_bx | on_by)[:, :, None].repeat(3, axis=2)
    mismatch = got != ref
    assert not np.any(mismatch & ~boundary), "mismatch away from sampling boundary"
    assert mismatch.mean() <= 0.25


def to_pil(a):
    return PIL.Image.fromarray(a, mode="RGB")


def from_pil(im):
    return np.asarray(im, dtype=np.uint8)


@pytest.mark.parametrize("v", [-0.3, -0.1, 0.05, 0.3])
def test_shear_x(v):
    a = rand_img()
    ref = from_pil(to_pil(a).transform((32, 32), PIL.Image.AFFINE, (1, v, 0, 0, 1, 0)))
    got = cpu_exec.affine_nearest(a, (1, v, 0, 0, 1, 0))
    assert_affine_close(got, ref, (1, v, 0, 0, 1, 0))


@pytest.mark.parametrize("v", [-0.3, 0.2])
def test_shear_y(v):
    a = rand_img()
    ref = from_pil(to_pil(a).transform((32, 32), PIL.Image.AFFINE, (1, 0, 0, v, 1, 0)))
    got = cpu_exec.affine_nearest(a, (1, 0, 0, v, 1, 0))
    assert_affine_close(got, ref, (1, 0, 0, v, 1, 0))


@pytest.mark.parametrize("v", [-10.3, -3, 0.0, 7.9])
def test_translate(v):
    a = rand_img()
    ref = from_pil(to_pil(a).transform((32, 32), PIL.Image.AFFINE, (1, 0, v, 0, 1, 0)))
    got = cpu_exec.affine_nearest(a, (1, 0, v, 0, 1, 0))
    assert_affine_close(got, ref, (1, 0, v, 0, 1, 0))


@pytest.mark.parametrize("deg", [-30, -13.7, 5.0, 30])
def test_rotate(deg):
    a = rand_img()
    ref = from_pil(to_pil(a).rotate(deg))
    m = _rotate_matrix(32, 32, deg)
    got = cpu_exec.affine_nearest(a, m)
    assert_affine_close(got, ref, m)


def test_autocontrast():
    # use a low-dynamic-range image so autocontrast actually stretches
    a = (rand_img() // 3 + 40).astype(np.uint8)
    ref = from_pil(PIL.ImageOps.autocontrast(to_pil(a)))
    got = cpu_exec.autocontrast(a)
    np.testing.assert_array_equal(got, ref)


def test_invert():
    a = rand_img()
    np.testing.assert_array_equal(cpu_exec.invert(a), from_pil(PIL.ImageOps.invert(to_pil(a))))


def test_equalize():
    a = rand_img()
    ref = from_pil(PIL.ImageOps.equalize(to_pil(a)))
    np.testing.assert_array_equal(cpu_exec.equalize(a), ref)


def test_equalize_skewed():
    a = (rand_img() // 7).astype(np.uint8)
    ref = from_pil(PIL.ImageOps.equalize(to_pil(a)))
    np.testing.assert_array_equal(cpu_exec.equalize(a), ref)


@pytest.mark.parametrize("th", [0, 64, 128.7, 255, 256])
def test_solarize(th):
    a = rand_img()
    ref = from_pil(PIL.ImageOps.solarize(to_pil(a), th))
    np.testing.assert_array_equal(cpu_exec.solarize(a, th), ref)


@pytest.mark.parametrize("bits", [1, 4, 5, 7, 8])
def test_posterize(bits):
    a = rand_img()
    ref = from_pil(PIL.ImageOps.posterize(to_pil(a), bits))
    np.testing.assert_array_equal(cpu_exec.posterize(a, bits), ref)


@pytest.mark.parametrize("f", [0.1, 0.55, 1.0, 1.9])
def test_contrast(f):
    a = rand_img()
    ref = from_pil(PIL.ImageEnhance.Contrast(to_pil(a)).enhance(f))
    got = cpu_exec.contrast(a, f)
    assert np.abs(got.astype(int) - ref.astype(int)).max() <= 1


@pytest.mark.parametrize("f", [0.1, 0.55, 1.0, 1.9])
def test_color(f):
    a = rand_img()
    ref = from_pil(PIL.ImageEnhance.Color(to_pil(a)).enhance(f))
    got = cpu_exec.color(a, f)
    assert np.abs(got.astype(int) - ref.astype(int)).max() <= 1


@pytest.mark.parametrize("f", [0.1, 0.55, 1.0, 1.9])
def test_brightness(f):
    a = rand_img()
    ref = from_pil(PIL.ImageEnhance.Brightness(to_pil(a)).enhance(f))
    got = cpu_exec.brightness(a, f)
    assert np.abs(got.astype(int) - ref.astype(int)).max() <= 1


@pytest.mark.parametrize("f", [0.1, 0.55, 1.0, 1.9])
def test_sharpness(f):
    a = rand_img()
    ref = from_pil(PIL.ImageEnhance.Sharpness(to_pil(a)).enhance(f))
    got = cpu_exec.sharpness(a, f)
    assert np.abs(got.astype(int) - ref.astype(int)).max() <= 1


def test_cutout_fill():
    a = rand_img()
    im = to_pil(a).copy()
    PIL.ImageDraw.Draw(im).rectangle((5, 7, 20, 22), (125, 123, 114))
    got = cpu_exec.cutout_fill(a, 5, 7, 20, 22)
    np.testing.assert_array_equal(got, from_pil(im))


def test_hflip():
    a = rand_img()
    np.testing.assert_array_equal(cpu_exec.hflip(a), from_pil(PIL.ImageOps.mirror(to_pil(a))))


@pytest.mark.parametrize("alpha", [0.0, 0.13, 0.4])
def test_sample_pairing(alpha):
    """SamplePairing = PIL Image.blend (reference augmentations.py:147-152)."""
    a, b = rand_img(), rand_img()
    ref = from_pil(PIL.Image.blend(to_pil(a), to_pil(b), alpha))
    got = cpu_exec.sample_pairing(a, b, alpha)
    assert np.abs(got.astype(int) - ref.astype(int)).max() <= 1
