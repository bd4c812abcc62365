// Baseline JPEG decoder (host C++) — the ImageRecordIter image path
// (reference parity: src/io/iter_image_recordio_2.cc decoded JPEG via
// OpenCV; this environment ships no OpenCV/libjpeg headers, so the
// decoder is implemented directly).  Scope: baseline sequential DCT
// (SOF0), 8-bit samples, 1 or 3 components, 4:4:4 / 4:2:0 / 4:2:2 / 4:1:1
// chroma subsampling, restart markers.  Progressive (SOF2) and arithmetic
// coding are rejected with a clear error.
//
// Output: packed RGB (or replicated grayscale) uint8, row-major HWC.
#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace mxa_jpeg {

typedef uint8_t u8;
typedef uint16_t u16;
typedef uint32_t u32;

struct Huff {
  // canonical Huffman: per length 1..16 the first code and value offset
  int mincode[17], maxcode[18], valptr[17];
  u8 vals[256];
  bool present = false;
};

struct Component {
  int id, h, v, tq;  // sampling factors, quant table
  int td, ta;        // huffman tables (DC, AC)
  int dcpred;
  std::vector<int> data;  // per-component plane (blocks_w*8 x blocks_h*8)
  int pw, ph;             // plane dims (padded to MCU)
};

struct Reader {
  const u8* p;
  const u8* end;
  explicit Reader(const u8* b, size_t n) : p(b), end(b + n) {}
  u8 byte() {
    if (p >= end) throw std::runtime_error("jpeg: truncated stream");
    return *p++;
  }
  u16 word() {
    u16 hi = byte();
    return (u16)((hi << 8) | byte());
  }
  void skip(int n) {
    if (p + n > end) throw std::runtime_error("jpeg: truncated segment");
    p += n;
  }
};

// entropy-coded bit reader with 0xFF00 stuffing + restart-marker handling
struct BitReader {
  Reader& r;
  u32 buf = 0;
  int nbits = 0;
  bool marker_hit = false;
  explicit BitReader(Reader& rr) : r(rr) {}
  void reset() {
    buf = 0;
    nbits = 0;
    marker_hit = false;
  }
  int bit() {
    if (nbits == 0) {
      if (marker_hit) return 0;  // pad past a marker (spec behavior)
      u8 b = r.byte();
      if (b == 0xFF) {
        u8 b2 = r.byte();
        if (b2 == 0x00) {
          // stuffed FF
        } else {
          // a marker (RSTn or EOI): un-consume and pad with zeros
          r.p -= 2;
          marker_hit = true;
          return 0;
        }
      }
      buf = b;
      nbits = 8;
    }
    --nbits;
    return (buf >> nbits) & 1;
  }
  int bits(int n) {
    int v = 0;
    for (int i = 0; i < n; ++i) v = (v << 1) | bit();
    return v;
  }
};

static int huff_decode(BitReader& br, const Huff& h) {
  int code = 0;
  for (int l = 1; l <= 16; ++l) {
    code = (code << 1) | br.bit();
    if (h.maxcode[l] >= 0 && code <= h.maxcode[l] && code >= h.mincode[l])
      return h.vals[h.valptr[l] + code - h.mincode[l]];
  }
  throw std::runtime_error("jpeg: bad huffman code");
}

static int receive_extend(BitReader& br, int s) {
  if (s == 0) return 0;
  int v = br.bits(s);
  if (v < (1 << (s - 1))) v += ((-1) << s) + 1;
  return v;
}

static const int kZigzag[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

// separable float IDCT (exactness is bounded by the quantization anyway)
static void idct8(float* v, int stride) {
  static const float c[8] = {0.70710678f, 0.98078528f, 0.92387953f,
                             0.83146961f, 0.70710678f, 0.55557023f,
                             0.38268343f, 0.19509032f};
  float tmp[8];
  for (int x = 0; x < 8; ++x) {
    float s = 0;
    for (int u = 0; u < 8; ++u) {
      float cu = u == 0 ? 0.70710678f : 1.0f;
      s += cu * v[u * stride] *
           cosf((2 * x + 1) * u * 3.14159265358979f / 16.0f);
    }
    tmp[x] = s * 0.5f;
  }
  (void)c;
  for (int x = 0; x < 8; ++x) v[x * stride] = tmp[x];
}

static void idct_block(const int* coef, const u16* qt, int* out) {
  float b[64];
  for (int i = 0; i < 64; ++i) b[i] = (float)coef[i] * qt[i];
  for (int col = 0; col < 8; ++col) idct8(&b[col], 8);
  for (int row = 0; row < 8; ++row) idct8(&b[row * 8], 1);
  for (int i = 0; i < 64; ++i) {
    int v = (int)lrintf(b[i]) + 128;
    out[i] = v < 0 ? 0 : (v > 255 ? 255 : v);
  }
}

struct Decoder {
  u16 qtab[4][64] = {};
  Huff hdc[4], hac[4];
  Component comp[3];
  int ncomp = 0, W = 0, H = 0, restart_interval = 0;
  int hmax = 1, vmax = 1;

  void parse_dqt(Reader& r, int len) {
    int remain = len - 2;
    while (remain > 0) {
      u8 pq_tq = r.byte();
      int pq = pq_tq >> 4, tq = pq_tq & 15;
      if (tq > 3) throw std::runtime_error("jpeg: bad DQT id");
      for (int i = 0; i < 64; ++i)
        qtab[tq][kZigzag[i]] = pq ? r.word() : r.byte();
      remain -= 1 + (pq ? 128 : 64);
    }
  }

  void parse_dht(Reader& r, int len) {
    int remain = len - 2;
    while (remain > 0) {
      u8 tc_th = r.byte();
      int tc = tc_th >> 4, th = tc_th & 15;
      if (th > 3) throw std::runtime_error("jpeg: bad DHT id");
      Huff& h = tc ? hac[th] : hdc[th];
      int counts[17], total = 0;
      for (int l = 1; l <= 16; ++l) {
        counts[l] = r.byte();
        total += counts[l];
      }
      if (total > 256) throw std::runtime_error("jpeg: bad DHT counts");
      for (int i = 0; i < total; ++i) h.vals[i] = r.byte();
      int code = 0, k = 0;
      for (int l = 1; l <= 16; ++l) {
        h.valptr[l] = k;
        h.mincode[l] = code;
        code += counts[l];
        k += counts[l];
        h.maxcode[l] = counts[l] ? code - 1 : -1;
        code <<= 1;
      }
      h.present = true;
      remain -= 17 + total;
    }
  }

  void parse_sof0(Reader& r) {
    int prec = r.byte();
    if (prec != 8) throw std::runtime_error("jpeg: only 8-bit supported");
    H = r.word();
    W = r.word();
    ncomp = r.byte();
    if (ncomp != 1 && ncomp != 3)
      throw std::runtime_error("jpeg: 1 or 3 components supported");
    for (int i = 0; i < ncomp; ++i) {
      comp[i].id = r.byte();
      u8 hv = r.byte();
      comp[i].h = hv >> 4;
      comp[i].v = hv & 15;
      comp[i].tq = r.byte();
      if (comp[i].h < 1 || comp[i].h > 4 || comp[i].v < 1 || comp[i].v > 4)
        throw std::runtime_error("jpeg: bad sampling factors");
      hmax = std::max(hmax, comp[i].h);
      vmax = std::max(vmax, comp[i].v);
    }
  }

  void decode_scan(Reader& r) {
    // SOS header
    int ns = r.byte();
    if (ns != ncomp)
      throw std::runtime_error("jpeg: interleaved full scan expected");
    for (int i = 0; i < ns; ++i) {
      int id = r.byte();
      u8 tdta = r.byte();
      for (int c = 0; c < ncomp; ++c)
        if (comp[c].id == id) {
          comp[c].td = tdta >> 4;
          comp[c].ta = tdta & 15;
        }
    }
    r.skip(3);  // Ss, Se, Ah/Al (baseline: 0, 63, 0)

    int mcux = (W + 8 * hmax - 1) / (8 * hmax);
    int mcuy = (H + 8 * vmax - 1) / (8 * vmax);
    for (int c = 0; c < ncomp; ++c) {
      comp[c].pw = mcux * 8 * comp[c].h;
      comp[c].ph = mcuy * 8 * comp[c].v;
      comp[c].data.assign((size_t)comp[c].pw * comp[c].ph, 0);
      comp[c].dcpred = 0;
    }

    BitReader br(r);
    int coef[64], pix[64];
    int mcu_count = 0;
    for (int my = 0; my < mcuy; ++my)
      for (int mx = 0; mx < mcux; ++mx) {
        if (restart_interval && mcu_count &&
            mcu_count % restart_interval == 0) {
          // expect RSTn marker
          br.reset();
          u8 ff = r.byte(), m = r.byte();
          if (ff != 0xFF || m < 0xD0 || m > 0xD7)
            throw std::runtime_error("jpeg: missing restart marker");
          for (int c = 0; c < ncomp; ++c) comp[c].dcpred = 0;
        }
        for (int c = 0; c < ncomp; ++c) {
          Component& co = comp[c];
          for (int by = 0; by < co.v; ++by)
            for (int bx = 0; bx < co.h; ++bx) {
              memset(coef, 0, sizeof(coef));
              // DC
              int s = huff_decode(br, hdc[co.td]);
              co.dcpred += receive_extend(br, s);
              coef[0] = co.dcpred;
              // AC
              for (int k = 1; k < 64;) {
                int rs = huff_decode(br, hac[co.ta]);
                int rrrr = rs >> 4, ssss = rs & 15;
                if (ssss == 0) {
                  if (rrrr == 15) {
                    k += 16;
                    continue;
                  }
                  break;  // EOB
                }
                k += rrrr;
                if (k > 63)
                  throw std::runtime_error("jpeg: AC index overflow");
                coef[kZigzag[k]] = receive_extend(br, ssss);
                ++k;
              }
              idct_block(coef, qtab[co.tq], pix);
              int x0 = (mx * co.h + bx) * 8, y0 = (my * co.v + by) * 8;
              for (int yy = 0; yy < 8; ++yy)
                for (int xx = 0; xx < 8; ++xx)
                  co.data[(size_t)(y0 + yy) * co.pw + x0 + xx] =
                      pix[yy * 8 + xx];
            }
        }
        ++mcu_count;
      }
  }

  // decoded planes -> packed RGB uint8 (nearest-neighbor chroma upsample)
  void to_rgb(std::vector<u8>& out) {
    out.resize((size_t)W * H * 3);
    if (ncomp == 1) {
      const Component& y = comp[0];
      for (int r = 0; r < H; ++r)
        for (int c = 0; c < W; ++c) {
          u8 v = (u8)y.data[(size_t)r * y.pw + c];
          u8* o = &out[((size_t)r * W + c) * 3];
          o[0] = o[1] = o[2] = v;
        }
      return;
    }
    const Component &Y = comp[0], &Cb = comp[1], &Cr = comp[2];
    for (int r = 0; r < H; ++r) {
      int rcb = r * Cb.v / vmax, rcr = r * Cr.v / vmax;
      for (int c = 0; c < W; ++c) {
        int yv = Y.data[(size_t)(r * Y.v / vmax) * Y.pw +
                        (size_t)(c * Y.h / hmax)];
        int cb = Cb.data[(size_t)rcb * Cb.pw + c * Cb.h / hmax] - 128;
        int cr = Cr.data[(size_t)rcr * Cr.pw + c * Cr.h / hmax] - 128;
        int R = (int)lrintf(yv + 1.402f * cr);
        int G = (int)lrintf(yv - 0.344136f * cb - 0.714136f * cr);
        int B = (int)lrintf(yv + 1.772f * cb);
        u8* o = &out[((size_t)r * W + c) * 3];
        o[0] = (u8)(R < 0 ? 0 : R > 255 ? 255 : R);
        o[1] = (u8)(G < 0 ? 0 : G > 255 ? 255 : G);
        o[2] = (u8)(B < 0 ? 0 : B > 255 ? 255 : B);
      }
    }
  }
};

// public entry: decode `buf` -> RGB bytes + dims.  Throws on non-baseline.
void decode(const u8* buf, size_t n, std::vector<u8>& rgb, int* w, int* h) {
  Reader r(buf, n);
  if (r.byte() != 0xFF || r.byte() != 0xD8)
    throw std::runtime_error("jpeg: missing SOI");
  Decoder d;
  bool sof_seen = false;
  for (;;) {
    u8 ff = r.byte();
    if (ff != 0xFF) continue;  // tolerate fill
    u8 m = r.byte();
    while (m == 0xFF) m = r.byte();
    if (m == 0xD8) continue;           // stray SOI
    if (m == 0xD9) break;              // EOI
    if (m >= 0xD0 && m <= 0xD7) continue;  // stray RST
    int len = r.word();
    const u8* seg_end = r.p + len - 2;
    switch (m) {
      case 0xDB:
        d.parse_dqt(r, len);
        break;
      case 0xC4:
        d.parse_dht(r, len);
        break;
      case 0xC0:  // baseline
      case 0xC1:  // extended sequential (same decode path)
        d.parse_sof0(r);
        sof_seen = true;
        break;
      case 0xC2:
        throw std::runtime_error(
            "jpeg: progressive (SOF2) not supported by the baseline "
            "decoder");
      case 0xC9:
      case 0xCA:
      case 0xCB:
        throw std::runtime_error("jpeg: arithmetic coding not supported");
      case 0xDD:
        d.restart_interval = r.word();
        break;
      case 0xDA:
        if (!sof_seen) throw std::runtime_error("jpeg: SOS before SOF");
        d.decode_scan(r);
        d.to_rgb(rgb);
        *w = d.W;
        *h = d.H;
        return;
      default:
        r.p = seg_end;  // skip APPn/COM/etc
        break;
    }
    if (r.p < seg_end) r.p = seg_end;
  }
  throw std::runtime_error("jpeg: no scan found");
}

}  // namespace mxa_jpeg
