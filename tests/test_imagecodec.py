"""In-tree image codec (csrc/imagecodec.cpp) vs from-scratch Python
encoders: PNG (zlib-compressed, all color types + all filter types), BMP,
and baseline JPEG produced by a scipy-DCT encoder (gray, 4:4:4, 4:2:0).
Host-only C++ — runs in CPU CI (reference stb_image analog,
src/data_loading/stb_image_impl.cpp)."""

import struct
import zlib

import numpy as np
import pytest
import torch

from tnn_amd import _C

try:
    ext = _C.ext_any()
except RuntimeError:  # pragma: no cover
    ext = None

pytestmark = pytest.mark.skipif(ext is None, reason="extension not built")


# ---------------------------------------------------------------------------
# PNG writer (zlib from python stdlib; filters applied per row)
# ---------------------------------------------------------------------------

def _png_chunk(tag, body):
    return (struct.pack(">I", len(body)) + tag + body
            + struct.pack(">I", zlib.crc32(tag + body)))


def write_png(img: np.ndarray, ctype: int, palette=None, filters=None):
    h, w = img.shape[:2]
    ch = 1 if img.ndim == 2 else img.shape[2]
    img2 = img.reshape(h, w * ch).astype(np.int32)
    raw = bytearray()
    prev = np.zeros(w * ch, dtype=np.int32)
    for y in range(h):
        f = (filters[y % len(filters)] if filters else 0)
        cur = img2[y]
        a = np.zeros_like(cur)
        a[ch:] = cur[:-ch]
        b = prev
        c = np.zeros_like(prev)
        c[ch:] = prev[:-ch]
        if f == 0:
            enc = cur
        elif f == 1:
            enc = cur - a
        elif f == 2:
            enc = cur - b
        elif f == 3:
            enc = cur - ((a + b) >> 1)
        else:
            p = a + b - c
            pa, pb, pc = np.abs(p - a), np.abs(p - b), np.abs(p - c)
            pred = np.where((pa <= pb) & (pa <= pc), a,
                            np.where(pb <= pc, b, c))
            enc = cur - pred
        raw.append(f)
        raw += (enc & 0xFF).astype(np.uint8).tobytes()
        prev = cur
    hdr = struct.pack(">IIBBBBB", w, h, 8, ctype, 0, 0, 0)
    out = b"\x89PNG\r\n\x1a\n" + _png_chunk(b"IHDR", hdr)
    if palette is not None:
        out += _png_chunk(b"PLTE", palette.astype(np.uint8).tobytes())
    out += _png_chunk(b"IDAT", zlib.compress(raw.__bytes__()
                                             if hasattr(raw, "__bytes__")
                                             else bytes(raw), 6))
    out += _png_chunk(b"IEND", b"")
    return out


@pytest.mark.parametrize("filters", [[0], [0, 1, 2, 3, 4]])
def test_png_rgb(filters):
    rng = np.random.default_rng(0)
    img = rng.integers(0, 256, (23, 17, 3), dtype=np.uint8)
    out = ext.decode_image(write_png(img, 2, filters=filters))
    assert out.shape == (23, 17, 3)
    assert np.array_equal(out.numpy(), img)


def test_png_gray_and_rgba():
    rng = np.random.default_rng(1)
    g = rng.integers(0, 256, (9, 31), dtype=np.uint8)
    out = ext.decode_image(write_png(g, 0, filters=[0, 4]))
    assert out.shape == (9, 31, 1)
    assert np.array_equal(out.numpy()[..., 0], g)
    rgba = rng.integers(0, 256, (12, 8, 4), dtype=np.uint8)
    out = ext.decode_image(write_png(rgba, 6, filters=[3, 1]))
    assert np.array_equal(out.numpy(), rgba)


def test_png_palette():
    rng = np.random.default_rng(2)
    pal = rng.integers(0, 256, (16, 3), dtype=np.uint8)
    idx = rng.integers(0, 16, (10, 14), dtype=np.uint8)
    out = ext.decode_image(write_png(idx, 3, palette=pal))
    assert out.shape == (10, 14, 3)
    assert np.array_equal(out.numpy(), pal[idx])


def test_png_large_dynamic_huffman():
    """A big low-entropy image forces dynamic-huffman + long matches."""
    y, x = np.mgrid[0:200, 0:300]
    img = ((y // 10 * 20 + x // 30 * 10) % 256).astype(np.uint8)
    img3 = np.stack([img, 255 - img, img ^ 0x55], axis=-1)
    out = ext.decode_image(write_png(img3, 2, filters=[2]))
    assert np.array_equal(out.numpy(), img3)


# ---------------------------------------------------------------------------
# BMP writer
# ---------------------------------------------------------------------------

def write_bmp(img: np.ndarray, bpp=24):
    h, w = img.shape[:2]
    ch = bpp // 8
    rowsz = (w * ch + 3) & ~3
    data = bytearray()
    for y in range(h - 1, -1, -1):  # bottom-up
        row = bytearray()
        for x in range(w):
            r, g, b = img[y, x][:3]
            row += bytes([b, g, r] + ([255] if ch == 4 else []))
        row += b"\0" * (rowsz - len(row))
        data += row
    off = 54
    hdr = (b"BM" + struct.pack("<IHHI", off + len(data), 0, 0, off)
           + struct.pack("<IiiHHIIiiII", 40, w, h, 1, bpp, 0, len(data),
                         2835, 2835, 0, 0))
    return hdr + bytes(data)


@pytest.mark.parametrize("bpp", [24, 32])
def test_bmp(bpp):
    rng = np.random.default_rng(3)
    img = rng.integers(0, 256, (11, 7, 3), dtype=np.uint8)
    out = ext.decode_image(write_bmp(img, bpp))
    assert np.array_equal(out.numpy(), img)


# ---------------------------------------------------------------------------
# Baseline JPEG encoder (scipy DCT; simple fixed-length-8 huffman tables)
# ---------------------------------------------------------------------------

ZZ = [0, 1, 8, 16, 9, 2, 3, 10, 17, 24, 32, 25, 18, 11, 4, 5,
      12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6, 7, 14, 21, 28,
      35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
      58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63]

DC_SYMS = list(range(12))
AC_SYMS = [0x00, 0xF0] + [(r << 4) | s for r in range(16)
                          for s in range(1, 11)]


class _Huf:
    """All codes length 8, canonical (code i = i); valid while < 255."""

    def __init__(self, symbols):
        assert len(symbols) < 255
        self.symbols = symbols
        self.codes = {s: i for i, s in enumerate(symbols)}

    def dht(self, tc, th):
        counts = [0] * 16
        counts[7] = len(self.symbols)
        body = bytes([(tc << 4) | th]) + bytes(counts) + bytes(self.symbols)
        return b"\xff\xc4" + struct.pack(">H", len(body) + 2) + body


class _BW:
    def __init__(self):
        self.out = bytearray()
        self.acc, self.n = 0, 0

    def put(self, val, nbits):
        for i in range(nbits - 1, -1, -1):
            self.acc = (self.acc << 1) | ((val >> i) & 1)
            self.n += 1
            if self.n == 8:
                self.out.append(self.acc)
                if self.acc == 0xFF:
                    self.out.append(0)
                self.acc, self.n = 0, 0

    def flush(self):
        while self.n:
            self.put(1, 1)


def _cat(v):
    return int(abs(v)).bit_length()


def _emit_val(bw, v, c):
    bw.put(v if v >= 0 else v + (1 << c) - 1, c)


def _encode_block(bw, block, qnat, pred, dch, ach):
    from scipy.fft import dctn
    coef = np.round(dctn(block.astype(np.float64) - 128.0, type=2,
                         norm="ortho") / qnat).astype(int)
    zz = [int(coef.flat[ZZ[k]]) for k in range(64)]
    diff = zz[0] - pred
    c = _cat(diff)
    bw.put(dch.codes[c], 8)
    if c:
        _emit_val(bw, diff, c)
    run = 0
    for k in range(1, 64):
        if zz[k] == 0:
            run += 1
            continue
        while run > 15:
            bw.put(ach.codes[0xF0], 8)
            run -= 16
        s = _cat(zz[k])
        bw.put(ach.codes[(run << 4) | s], 8)
        _emit_val(bw, zz[k], s)
        run = 0
    if run:
        bw.put(ach.codes[0x00], 8)
    return zz[0]


def _blocks_of(plane, by, bx):
    """8x8 block at block-coords (by,bx), edge-replicated."""
    h, w = plane.shape
    ys = np.clip(np.arange(by * 8, by * 8 + 8), 0, h - 1)
    xs = np.clip(np.arange(bx * 8, bx * 8 + 8), 0, w - 1)
    return plane[np.ix_(ys, xs)]


def write_jpeg(img: np.ndarray, subsample="444", quant=4):
    """img uint8 [H,W] gray or [H,W,3] RGB."""
    h, w = img.shape[:2]
    gray = img.ndim == 2
    qnat = np.full((8, 8), quant, dtype=np.float64)
    dqt_body = bytes([0]) + bytes(int(qnat.flat[ZZ[i]]) for i in range(64))
    dch, ach = _Huf(DC_SYMS), _Huf(AC_SYMS)

    if gray:
        comps = [(1, 1, 1)]
        planes = [img.astype(np.float64)]
    else:
        f = img.astype(np.float64)
        y = 0.299 * f[..., 0] + 0.587 * f[..., 1] + 0.114 * f[..., 2]
        cb = -0.168736 * f[..., 0] - 0.331264 * f[..., 1] + 0.5 * f[..., 2] + 128
        cr = 0.5 * f[..., 0] - 0.418688 * f[..., 1] - 0.081312 * f[..., 2] + 128
        if subsample == "420":
            comps = [(1, 2, 2), (2, 1, 1), (3, 1, 1)]
            he, we = (h + 1) & ~1, (w + 1) & ~1
            cbp = np.pad(cb, ((0, he - h), (0, we - w)), mode="edge")
            crp = np.pad(cr, ((0, he - h), (0, we - w)), mode="edge")
            cb2 = cbp.reshape(he // 2, 2, we // 2, 2).mean(axis=(1, 3))
            cr2 = crp.reshape(he // 2, 2, we // 2, 2).mean(axis=(1, 3))
            planes = [y, cb2, cr2]
        else:
            comps = [(1, 1, 1), (2, 1, 1), (3, 1, 1)]
            planes = [y, cb, cr]

    hmax = max(c[1] for c in comps)
    vmax = max(c[2] for c in comps)
    mcux = (w + 8 * hmax - 1) // (8 * hmax)
    mcuy = (h + 8 * vmax - 1) // (8 * vmax)

    out = bytearray(b"\xff\xd8")
    out += b"\xff\xdb" + struct.pack(">H", len(dqt_body) + 2) + dqt_body
    sof = bytes([8]) + struct.pack(">HH", h, w) + bytes([len(comps)])
    for cid, ch_, cv in comps:
        sof += bytes([cid, (ch_ << 4) | cv, 0])
    out += b"\xff\xc0" + struct.pack(">H", len(sof) + 2) + sof
    out += dch.dht(0, 0) + ach.dht(1, 0)
    sos = bytes([len(comps)])
    for cid, _, _ in comps:
        sos += bytes([cid, 0x00])
    sos += bytes([0, 63, 0])
    out += b"\xff\xda" + struct.pack(">H", len(sos) + 2) + sos

    bw = _BW()
    preds = [0] * len(comps)
    for my in range(mcuy):
        for mx in range(mcux):
            for ci, (cid, ch_, cv) in enumerate(comps):
                for v in range(cv):
                    for hh in range(ch_):
                        blk = _blocks_of(planes[ci], my * cv + v,
                                         mx * ch_ + hh)
                        preds[ci] = _encode_block(bw, blk, qnat, preds[ci],
                                                  dch, ach)
    bw.flush()
    out += bw.out + b"\xff\xd9"
    return bytes(out)


def test_jpeg_gray():
    rng = np.random.default_rng(4)
    base = rng.integers(40, 216, (20, 12), dtype=np.uint8).astype(np.float64)
    # smooth a little so quantization error stays local
    img = np.clip(base, 0, 255).astype(np.uint8)
    out = ext.decode_image(write_jpeg(img))
    assert out.shape == (20, 12, 1)
    err = np.abs(out.numpy()[..., 0].astype(int) - img.astype(int)).max()
    assert err <= 16, err


@pytest.mark.parametrize("subsample", ["444", "420"])
def test_jpeg_color(subsample):
    rng = np.random.default_rng(5)
    # smooth gradient + mild noise (chroma subsampling is lossy on edges)
    y, x = np.mgrid[0:32, 0:32]
    img = np.stack([(y * 6) % 230 + 10, (x * 5) % 220 + 15,
                    ((x + y) * 3) % 200 + 25], axis=-1).astype(np.uint8)
    out = ext.decode_image(write_jpeg(img, subsample=subsample))
    assert out.shape == (32, 32, 3)
    err = np.abs(out.numpy().astype(int) - img.astype(int))
    tol = 24 if subsample == "420" else 16
    assert np.percentile(err, 99) <= tol, (err.max(), np.percentile(err, 99))


def test_jpeg_solid_color_exact():
    img = np.full((16, 16, 3), 0, dtype=np.uint8)
    img[..., 0] = 180
    img[..., 1] = 90
    img[..., 2] = 40
    out = ext.decode_image(write_jpeg(img, subsample="420"))
    err = np.abs(out.numpy().astype(int) - img.astype(int)).max()
    assert err <= 6, err


def test_progressive_rejected():
    # minimal stream with SOF2 marker must raise, not crash
    bad = (b"\xff\xd8\xff\xc2" + struct.pack(">H", 11)
           + bytes([8]) + struct.pack(">HH", 8, 8) + bytes([1, 1, 0x11, 0]))
    with pytest.raises(RuntimeError, match="progressive"):
        ext.decode_image(bad)


def test_loader_integration(tmp_path):
    """Raw class-directory dataset loads end to end via the codec."""
    from tnn_amd.data.imageio import load_image_dir
    rng = np.random.default_rng(6)
    for ci, cls in enumerate(["n01", "n02"]):
        d = tmp_path / "train" / cls / "images"
        d.mkdir(parents=True)
        for i in range(3):
            img = rng.integers(0, 256, (16, 16, 3), dtype=np.uint8)
            (d / f"{cls}_{i}.png").write_bytes(write_png(img, 2))
    x, y = load_image_dir(str(tmp_path / "train"), size=16)
    assert x.shape == (6, 16, 16, 3)
    assert y.tolist() == [0, 0, 0, 1, 1, 1]
