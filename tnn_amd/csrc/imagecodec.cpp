// In-tree image decoding: baseline JPEG, PNG (with a from-scratch
// DEFLATE inflater) and BMP — the loader-side equivalent of the
// reference's stb_image dependency (src/data_loading/stb_image_impl.cpp),
// written from scratch so Tiny-ImageNet / ImageNet-100 load from the raw
// datasets without an offline preprocessing step.
//
// Host-only code (no HIP): decoding happens on CPU in the data loader,
// exactly as in the reference. Scope: baseline (sequential DCT) JPEG with
// 4:4:4/4:2:2/4:2:0 subsampling and restart markers; PNG bit-depth 8,
// color types gray/RGB/palette/gray-alpha/RGBA, non-interlaced;
// uncompressed 24/32-bit BMP. Progressive JPEG and Adam7 PNG are
// rejected with a clear error.

#include <torch/extension.h>

#include <cmath>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace tnn {
namespace imgcodec {

[[noreturn]] static void fail(const std::string& msg) {
  throw std::runtime_error("image decode: " + msg);
}

// ===========================================================================
// Baseline JPEG
// ===========================================================================
namespace jpg {

static const uint8_t ZZ[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

struct Huff {
  int mincode[17], maxcode[17], valptr[17];
  uint8_t vals[256];
  bool present = false;
  void build(const uint8_t counts[16], const uint8_t* symbols, int nsym) {
    int code = 0, k = 0;
    for (int l = 1; l <= 16; ++l) {
      valptr[l] = k;
      mincode[l] = code;
      if (counts[l - 1]) {
        for (int i = 0; i < counts[l - 1]; ++i, ++k) {
          if (k >= nsym || k >= 256) fail("jpeg: huffman overflow");
          vals[k] = symbols[k];
        }
        code += counts[l - 1];
        maxcode[l] = code - 1;
      } else {
        maxcode[l] = -1;
      }
      code <<= 1;
    }
    present = true;
  }
};

struct BitStream {
  const uint8_t* d;
  size_t n, p;
  uint32_t buf = 0;
  int cnt = 0;
  bool marker_hit = false;

  int bit() {
    if (cnt == 0) {
      if (p >= n) {
        marker_hit = true;
        return 0;
      }
      uint8_t b = d[p];
      if (b == 0xFF) {
        uint8_t m = (p + 1 < n) ? d[p + 1] : 0xD9;
        if (m == 0x00) {
          p += 2;  // stuffed 0xFF data byte
        } else {
          marker_hit = true;  // real marker: stop, handled by caller
          return 0;
        }
      } else {
        p += 1;
      }
      buf = b;
      cnt = 8;
    }
    --cnt;
    return (buf >> cnt) & 1;
  }
  int bits(int k) {
    int v = 0;
    while (k--) v = (v << 1) | bit();
    return v;
  }
  void byte_align() {
    cnt = 0;
    marker_hit = false;
  }
};

static int decode_huff(BitStream& bs, const Huff& h) {
  int code = 0;
  for (int l = 1; l <= 16; ++l) {
    code = (code << 1) | bs.bit();
    if (bs.marker_hit) fail("jpeg: bitstream ended inside huffman code");
    if (h.maxcode[l] >= 0 && code <= h.maxcode[l])
      return h.vals[h.valptr[l] + code - h.mincode[l]];
  }
  fail("jpeg: invalid huffman code");
}

static int extend(int v, int t) {
  return (t && v < (1 << (t - 1))) ? v - (1 << t) + 1 : v;
}

struct Comp {
  int id = 0, h = 1, v = 1, tq = 0, td = 0, ta = 0;
  int dcpred = 0;
  int pw = 0, ph = 0;  // padded plane dims (multiples of 8*h/8*v per MCU)
  std::vector<uint8_t> plane;
};

struct Decoder {
  const uint8_t* d;
  size_t n;
  uint16_t qt[4][64] = {};
  Huff hdc[4], hac[4];
  Comp comp[3];
  int ncomp = 0, W = 0, H = 0, hmax = 1, vmax = 1, restart = 0;
  float ctab[8][8];

  Decoder(const uint8_t* data, size_t len) : d(data), n(len) {
    for (int u = 0; u < 8; ++u)
      for (int x = 0; x < 8; ++x)
        ctab[u][x] = 0.5f * (u == 0 ? (float)M_SQRT1_2 : 1.0f) *
                     std::cos((2 * x + 1) * u * (float)M_PI / 16.0f);
  }

  uint16_t u16(size_t p) { return (uint16_t)((d[p] << 8) | d[p + 1]); }

  void idct_block(const int* in, uint8_t* out, int stride) {
    float tmp[64];
    for (int v = 0; v < 8; ++v)
      for (int x = 0; x < 8; ++x) {
        float s = 0.0f;
        for (int u = 0; u < 8; ++u) s += ctab[u][x] * in[v * 8 + u];
        tmp[v * 8 + x] = s;
      }
    for (int y = 0; y < 8; ++y)
      for (int x = 0; x < 8; ++x) {
        float s = 0.0f;
        for (int v = 0; v < 8; ++v) s += ctab[v][y] * tmp[v * 8 + x];
        int pv = (int)std::lround(s) + 128;
        out[y * stride + x] = (uint8_t)(pv < 0 ? 0 : (pv > 255 ? 255 : pv));
      }
  }

  void decode_block(BitStream& bs, Comp& c, int* blk) {
    std::memset(blk, 0, 64 * sizeof(int));
    const Huff& dc = hdc[c.td];
    const Huff& ac = hac[c.ta];
    const uint16_t* q = qt[c.tq];
    int t = decode_huff(bs, dc);
    int diff = t ? extend(bs.bits(t), t) : 0;
    c.dcpred += diff;
    blk[0] = c.dcpred * q[0];
    int k = 1;
    while (k < 64) {
      int rs = decode_huff(bs, ac);
      int r = rs >> 4, s = rs & 15;
      if (s == 0) {
        if (r == 15) {
          k += 16;
          continue;
        }
        break;  // EOB
      }
      k += r;
      if (k > 63) fail("jpeg: AC index overflow");
      blk[ZZ[k]] = extend(bs.bits(s), s) * q[k];
      ++k;
    }
  }

  at::Tensor run() {
    if (n < 4 || d[0] != 0xFF || d[1] != 0xD8) fail("jpeg: no SOI");
    size_t p = 2;
    bool sof_seen = false;
    while (p + 4 <= n) {
      if (d[p] != 0xFF) fail("jpeg: marker sync lost");
      uint8_t m = d[p + 1];
      p += 2;
      if (m == 0xD8 || (m >= 0xD0 && m <= 0xD7) || m == 0x01) continue;
      if (m == 0xD9) break;
      size_t len = u16(p);
      if (p + len > n) fail("jpeg: truncated segment");
      size_t seg = p + 2, seg_end = p + len;
      switch (m) {
        case 0xDB:  // DQT
          while (seg < seg_end) {
            int pq = d[seg] >> 4, tq = d[seg] & 15;
            ++seg;
            for (int i = 0; i < 64; ++i)
              if (pq) {
                qt[tq][i] = u16(seg);
                seg += 2;
              } else {
                qt[tq][i] = d[seg++];
              }
          }
          break;
        case 0xC4:  // DHT
          while (seg < seg_end) {
            int tc = d[seg] >> 4, th = d[seg] & 15;
            ++seg;
            uint8_t counts[16];
            int nsym = 0;
            for (int i = 0; i < 16; ++i) nsym += (counts[i] = d[seg + i]);
            seg += 16;
            (tc ? hac[th] : hdc[th]).build(counts, d + seg, nsym);
            seg += nsym;
          }
          break;
        case 0xC0:
        case 0xC1: {  // SOF0/1 baseline
          if (d[seg] != 8) fail("jpeg: only 8-bit precision supported");
          H = u16(seg + 1);
          W = u16(seg + 3);
          ncomp = d[seg + 5];
          if (ncomp != 1 && ncomp != 3) fail("jpeg: 1 or 3 components only");
          for (int i = 0; i < ncomp; ++i) {
            comp[i].id = d[seg + 6 + 3 * i];
            comp[i].h = d[seg + 7 + 3 * i] >> 4;
            comp[i].v = d[seg + 7 + 3 * i] & 15;
            comp[i].tq = d[seg + 8 + 3 * i];
            if (comp[i].h < 1 || comp[i].h > 2 || comp[i].v < 1 ||
                comp[i].v > 2)
              fail("jpeg: sampling factors beyond 2 unsupported");
            hmax = std::max(hmax, comp[i].h);
            vmax = std::max(vmax, comp[i].v);
          }
          sof_seen = true;
          break;
        }
        case 0xC2:
          fail("jpeg: progressive JPEG not supported (baseline only)");
        case 0xDD:  // DRI
          restart = u16(seg);
          break;
        case 0xDA: {  // SOS
          if (!sof_seen) fail("jpeg: SOS before SOF");
          int ns = d[seg];
          ++seg;
          for (int i = 0; i < ns; ++i) {
            int id = d[seg];
            for (int c = 0; c < ncomp; ++c)
              if (comp[c].id == id) {
                comp[c].td = d[seg + 1] >> 4;
                comp[c].ta = d[seg + 1] & 15;
              }
            seg += 2;
          }
          seg += 3;  // Ss/Se/AhAl
          return scan(seg);
        }
        default:
          break;  // APPn, COM, ...
      }
      p += len;
    }
    fail("jpeg: no scan found");
  }

  at::Tensor scan(size_t p) {
    const int mcux = (W + 8 * hmax - 1) / (8 * hmax);
    const int mcuy = (H + 8 * vmax - 1) / (8 * vmax);
    for (int c = 0; c < ncomp; ++c) {
      comp[c].pw = mcux * comp[c].h * 8;
      comp[c].ph = mcuy * comp[c].v * 8;
      comp[c].plane.assign((size_t)comp[c].pw * comp[c].ph, 0);
      comp[c].dcpred = 0;
    }
    BitStream bs{d, n, p};
    int blk[64];
    int mcu_count = 0;
    for (int my = 0; my < mcuy; ++my)
      for (int mx = 0; mx < mcux; ++mx) {
        if (restart && mcu_count && mcu_count % restart == 0) {
          bs.byte_align();
          if (bs.p + 1 < n && d[bs.p] == 0xFF && d[bs.p + 1] >= 0xD0 &&
              d[bs.p + 1] <= 0xD7)
            bs.p += 2;
          for (int c = 0; c < ncomp; ++c) comp[c].dcpred = 0;
        }
        for (int c = 0; c < ncomp; ++c)
          for (int v = 0; v < comp[c].v; ++v)
            for (int h = 0; h < comp[c].h; ++h) {
              decode_block(bs, comp[c], blk);
              uint8_t* out = comp[c].plane.data() +
                             (size_t)(my * comp[c].v + v) * 8 * comp[c].pw +
                             (size_t)(mx * comp[c].h + h) * 8;
              idct_block(blk, out, comp[c].pw);
            }
        ++mcu_count;
      }
    return color_convert();
  }

  at::Tensor color_convert() {
    const int C = ncomp == 1 ? 1 : 3;
    auto out = at::empty({H, W, C}, at::kByte);
    uint8_t* o = out.data_ptr<uint8_t>();
    if (ncomp == 1) {
      const Comp& Y = comp[0];
      for (int y = 0; y < H; ++y)
        std::memcpy(o + (size_t)y * W, Y.plane.data() + (size_t)y * Y.pw, W);
      return out;
    }
    for (int y = 0; y < H; ++y)
      for (int x = 0; x < W; ++x) {
        auto sample = [&](const Comp& c) -> int {
          int sy = y * c.v / vmax, sx = x * c.h / hmax;
          return c.plane[(size_t)sy * c.pw + sx];
        };
        float Y = (float)sample(comp[0]);
        float Cb = (float)sample(comp[1]) - 128.0f;
        float Cr = (float)sample(comp[2]) - 128.0f;
        auto clamp8 = [](float v) {
          int i = (int)std::lround(v);
          return (uint8_t)(i < 0 ? 0 : (i > 255 ? 255 : i));
        };
        uint8_t* px = o + ((size_t)y * W + x) * 3;
        px[0] = clamp8(Y + 1.402f * Cr);
        px[1] = clamp8(Y - 0.344136f * Cb - 0.714136f * Cr);
        px[2] = clamp8(Y + 1.772f * Cb);
      }
    return out;
  }
};

}  // namespace jpg

// ===========================================================================
// PNG + from-scratch DEFLATE
// ===========================================================================
namespace png {

struct BitIn {
  const uint8_t* d;
  size_t n, p = 0;
  uint32_t buf = 0;
  int cnt = 0;
  int bit() {
    if (cnt == 0) {
      if (p >= n) fail("deflate: out of data");
      buf = d[p++];
      cnt = 8;
    }
    int v = buf & 1;
    buf >>= 1;
    --cnt;
    return v;
  }
  int bits(int k) {  // LSB-first
    int v = 0;
    for (int i = 0; i < k; ++i) v |= bit() << i;
    return v;
  }
  void align() { cnt = 0; }
};

struct HuffD {
  // canonical decode from code lengths (codes are MSB-first)
  int counts[16] = {};
  std::vector<int> symbols;
  void build(const uint8_t* lens, int nsym) {
    symbols.clear();
    for (int i = 0; i < 16; ++i) counts[i] = 0;
    for (int i = 0; i < nsym; ++i) counts[lens[i]]++;
    counts[0] = 0;
    int offs[16] = {};
    for (int l = 1; l < 16; ++l) offs[l] = offs[l - 1] + counts[l - 1];
    symbols.resize(offs[15] + counts[15]);
    for (int i = 0; i < nsym; ++i)
      if (lens[i]) symbols[offs[lens[i]]++] = i;
  }
  int decode(BitIn& in) const {
    int code = 0, first = 0, index = 0;
    for (int l = 1; l < 16; ++l) {
      code |= in.bit();
      int cnt = counts[l];
      if (code - first < cnt) return symbols[index + (code - first)];
      index += cnt;
      first = (first + cnt) << 1;
      code <<= 1;
    }
    fail("deflate: invalid huffman code");
  }
};

static void inflate(const uint8_t* src, size_t n, std::vector<uint8_t>& out) {
  static const int LBASE[] = {3,  4,  5,  6,  7,  8,  9,  10, 11,  13,
                              15, 17, 19, 23, 27, 31, 35, 43, 51,  59,
                              67, 83, 99, 115, 131, 163, 195, 227, 258};
  static const int LEXT[] = {0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 2,
                             2, 3, 3, 3, 3, 4, 4, 4, 4, 5, 5, 5, 5, 0};
  static const int DBASE[] = {1,    2,    3,    4,    5,    7,     9,    13,
                              17,   25,   33,   49,   65,   97,    129,  193,
                              257,  385,  513,  769,  1025, 1537,  2049, 3073,
                              4097, 6145, 8193, 12289, 16385, 24577};
  static const int DEXT[] = {0, 0, 0, 0, 1, 1, 2, 2,  3,  3,  4,  4,  5, 5, 6,
                             6, 7, 7, 8, 8, 9, 9, 10, 10, 11, 11, 12, 12, 13, 13};
  BitIn in{src, n};
  HuffD lit, dist;
  for (;;) {
    int final = in.bit();
    int type = in.bits(2);
    if (type == 0) {
      in.align();
      if (in.p + 4 > n) fail("deflate: truncated stored block");
      int len = src[in.p] | (src[in.p + 1] << 8);
      in.p += 4;
      if (in.p + len > n) fail("deflate: truncated stored data");
      out.insert(out.end(), src + in.p, src + in.p + len);
      in.p += len;
    } else {
      if (type == 1) {
        uint8_t ll[288], dl[30];
        for (int i = 0; i < 144; ++i) ll[i] = 8;
        for (int i = 144; i < 256; ++i) ll[i] = 9;
        for (int i = 256; i < 280; ++i) ll[i] = 7;
        for (int i = 280; i < 288; ++i) ll[i] = 8;
        for (int i = 0; i < 30; ++i) dl[i] = 5;
        lit.build(ll, 288);
        dist.build(dl, 30);
      } else if (type == 2) {
        static const int ORD[] = {16, 17, 18, 0, 8,  7, 9,  6, 10, 5,
                                  11, 4,  12, 3, 13, 2, 14, 1, 15};
        int hlit = in.bits(5) + 257, hdist = in.bits(5) + 1,
            hclen = in.bits(4) + 4;
        uint8_t cl[19] = {};
        for (int i = 0; i < hclen; ++i) cl[ORD[i]] = (uint8_t)in.bits(3);
        HuffD clh;
        clh.build(cl, 19);
        uint8_t lens[288 + 32] = {};
        int i = 0;
        while (i < hlit + hdist) {
          int s = clh.decode(in);
          if (s < 16) lens[i++] = (uint8_t)s;
          else if (s == 16) {
            if (i == 0) fail("deflate: repeat with no previous length");
            int r = in.bits(2) + 3;
            while (r--) lens[i] = lens[i - 1], ++i;
          } else if (s == 17) {
            int r = in.bits(3) + 3;
            while (r--) lens[i++] = 0;
          } else {
            int r = in.bits(7) + 11;
            while (r--) lens[i++] = 0;
          }
        }
        lit.build(lens, hlit);
        dist.build(lens + hlit, hdist);
      } else {
        fail("deflate: bad block type");
      }
      for (;;) {
        int s = lit.decode(in);
        if (s < 256) {
          out.push_back((uint8_t)s);
        } else if (s == 256) {
          break;
        } else {
          s -= 257;
          if (s >= 29) fail("deflate: bad length symbol");
          int len = LBASE[s] + in.bits(LEXT[s]);
          int ds = dist.decode(in);
          if (ds >= 30) fail("deflate: bad distance symbol");
          size_t dd = (size_t)DBASE[ds] + in.bits(DEXT[ds]);
          if (dd > out.size()) fail("deflate: distance too far");
          size_t from = out.size() - dd;
          for (int k = 0; k < len; ++k) out.push_back(out[from + k]);
        }
      }
    }
    if (final) break;
  }
}

static int paeth(int a, int b, int c) {
  int pp = a + b - c, pa = std::abs(pp - a), pb = std::abs(pp - b),
      pc = std::abs(pp - c);
  return (pa <= pb && pa <= pc) ? a : (pb <= pc ? b : c);
}

static at::Tensor decode(const uint8_t* d, size_t n) {
  static const uint8_t SIG[8] = {137, 80, 78, 71, 13, 10, 26, 10};
  if (n < 8 || std::memcmp(d, SIG, 8) != 0) fail("png: bad signature");
  size_t p = 8;
  int W = 0, H = 0, depth = 0, ctype = 0, interlace = 0;
  std::vector<uint8_t> idat, pal;
  while (p + 8 <= n) {
    uint32_t len = (d[p] << 24) | (d[p + 1] << 16) | (d[p + 2] << 8) | d[p + 3];
    const char* tag = (const char*)(d + p + 4);
    const uint8_t* body = d + p + 8;
    if (p + 12 + len > n) fail("png: truncated chunk");
    if (!std::memcmp(tag, "IHDR", 4)) {
      W = (body[0] << 24) | (body[1] << 16) | (body[2] << 8) | body[3];
      H = (body[4] << 24) | (body[5] << 16) | (body[6] << 8) | body[7];
      depth = body[8];
      ctype = body[9];
      interlace = body[12];
      if (depth != 8) fail("png: only bit depth 8 supported");
      if (interlace) fail("png: Adam7 interlace not supported");
    } else if (!std::memcmp(tag, "PLTE", 4)) {
      pal.assign(body, body + len);
    } else if (!std::memcmp(tag, "IDAT", 4)) {
      idat.insert(idat.end(), body, body + len);
    } else if (!std::memcmp(tag, "IEND", 4)) {
      break;
    }
    p += 12 + len;
  }
  if (!W || !H || idat.size() < 2) fail("png: missing IHDR/IDAT");
  int ch;
  switch (ctype) {
    case 0: ch = 1; break;   // gray
    case 2: ch = 3; break;   // rgb
    case 3: ch = 1; break;   // palette indices
    case 4: ch = 2; break;   // gray+alpha
    case 6: ch = 4; break;   // rgba
    default: fail("png: bad color type");
  }
  // zlib wrapper: 2-byte header (+4 adler at end)
  std::vector<uint8_t> raw;
  raw.reserve((size_t)(ch * W + 1) * H);
  inflate(idat.data() + 2, idat.size() - 2, raw);
  const size_t stride = (size_t)ch * W;
  if (raw.size() < (stride + 1) * H) fail("png: inflated size mismatch");
  std::vector<uint8_t> img((size_t)H * stride);
  std::vector<uint8_t> prev(stride, 0);
  for (int y = 0; y < H; ++y) {
    const uint8_t* row = raw.data() + (size_t)y * (stride + 1);
    uint8_t filt = row[0];
    uint8_t* cur = img.data() + (size_t)y * stride;
    for (size_t x = 0; x < stride; ++x) {
      int rv = row[1 + x];
      int a = x >= (size_t)ch ? cur[x - ch] : 0;
      int b = prev[x];
      int c = x >= (size_t)ch ? prev[x - ch] : 0;
      switch (filt) {
        case 0: cur[x] = (uint8_t)rv; break;
        case 1: cur[x] = (uint8_t)(rv + a); break;
        case 2: cur[x] = (uint8_t)(rv + b); break;
        case 3: cur[x] = (uint8_t)(rv + ((a + b) >> 1)); break;
        case 4: cur[x] = (uint8_t)(rv + paeth(a, b, c)); break;
        default: fail("png: bad filter type");
      }
    }
    std::memcpy(prev.data(), cur, stride);
  }
  if (ctype == 3) {  // palette -> RGB
    if (pal.empty()) fail("png: palette image without PLTE");
    auto out = at::empty({H, W, 3}, at::kByte);
    uint8_t* o = out.data_ptr<uint8_t>();
    for (size_t i = 0; i < (size_t)H * W; ++i) {
      size_t idx = (size_t)img[i] * 3;
      if (idx + 2 >= pal.size()) fail("png: palette index out of range");
      o[i * 3] = pal[idx];
      o[i * 3 + 1] = pal[idx + 1];
      o[i * 3 + 2] = pal[idx + 2];
    }
    return out;
  }
  auto out = at::empty({H, W, ch}, at::kByte);
  std::memcpy(out.data_ptr<uint8_t>(), img.data(), img.size());
  return out;
}

}  // namespace png

// ===========================================================================
// BMP (uncompressed 24/32-bit)
// ===========================================================================
namespace bmp {

static at::Tensor decode(const uint8_t* d, size_t n) {
  if (n < 54 || d[0] != 'B' || d[1] != 'M') fail("bmp: bad signature");
  auto u32 = [&](size_t p) {
    return (uint32_t)(d[p] | (d[p + 1] << 8) | (d[p + 2] << 16) |
                      (d[p + 3] << 24));
  };
  uint32_t off = u32(10);
  int W = (int)u32(18);
  int Hs = (int)u32(22);
  int H = std::abs(Hs);
  int bpp = d[28] | (d[29] << 8);
  if (u32(30) != 0) fail("bmp: compressed BMP not supported");
  if (bpp != 24 && bpp != 32) fail("bmp: only 24/32-bit supported");
  const int ch = bpp / 8;
  const size_t rowsz = ((size_t)W * ch + 3) & ~3ull;
  if (off + rowsz * H > n) fail("bmp: truncated");
  auto out = at::empty({H, W, 3}, at::kByte);
  uint8_t* o = out.data_ptr<uint8_t>();
  for (int y = 0; y < H; ++y) {
    int sy = Hs > 0 ? H - 1 - y : y;  // bottom-up by default
    const uint8_t* row = d + off + rowsz * sy;
    for (int x = 0; x < W; ++x) {
      o[((size_t)y * W + x) * 3] = row[x * ch + 2];      // BGR -> RGB
      o[((size_t)y * W + x) * 3 + 1] = row[x * ch + 1];
      o[((size_t)y * W + x) * 3 + 2] = row[x * ch];
    }
  }
  return out;
}

}  // namespace bmp

at::Tensor decode_image(const py::bytes& data) {
  std::string s = data;  // copies; loader-side cost is fine
  const uint8_t* d = (const uint8_t*)s.data();
  size_t n = s.size();
  if (n >= 2 && d[0] == 0xFF && d[1] == 0xD8) {
    jpg::Decoder dec(d, n);
    return dec.run();
  }
  if (n >= 8 && d[0] == 137 && d[1] == 'P') return png::decode(d, n);
  if (n >= 2 && d[0] == 'B' && d[1] == 'M') return bmp::decode(d, n);
  fail("unknown image format (JPEG/PNG/BMP supported)");
}

}  // namespace imgcodec
}  // namespace tnn
