// jpegc.cpp — baseline-JPEG entropy decoder (SURVEY R15 full form).
//
// The serial half of JPEG decode (marker parse + Huffman) runs here on
// CPU with the GIL released; everything parallel — dequant, 8x8 IDCT,
// chroma upsample, YCbCr->RGB — runs as HIP kernels (jpeg_gpu.hip) on
// the ingest side stream. This module is torch-free so the shm-ring
// decode workers can import it cheaply.
//
// Scope: baseline sequential DCT (SOF0), 8-bit, 1 or 3 components,
// 4:4:4 / 4:2:2 / 4:2:0 sampling, restart intervals. Progressive
// (SOF2), arithmetic coding, 12-bit and CMYK raise ValueError — the
// caller falls back to the PIL pixel path for those.
//
// Output: per-component int16 DCT coefficient blocks, de-zigzagged to
// natural (row-major) order but still QUANTIZED; the dequant tables
// ship alongside so the GPU fuses dequant into the IDCT kernel.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

const uint8_t kZigzag[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

struct HuffTable {
  // canonical decode: mincode/maxcode/valptr per length, plus an 8-bit
  // fast lookup (symbol + length) for the common case
  int32_t mincode[17], maxcode[17], valptr[17];
  uint8_t vals[256];
  int32_t fast[256];  // (sym << 8) | len, or -1 (sym can be >= 128)
  bool present = false;

  void build(const uint8_t counts[16], const uint8_t* symbols, int nsym) {
    std::memcpy(vals, symbols, nsym);
    int code = 0, k = 0;
    for (int l = 1; l <= 16; ++l) {
      valptr[l] = k;
      mincode[l] = code;
      code += counts[l - 1];
      k += counts[l - 1];
      maxcode[l] = code - 1;
      code <<= 1;
    }
    for (int i = 0; i < 256; ++i) fast[i] = -1;
    // fill fast table for codes of length <= 8
    code = 0;
    k = 0;
    for (int l = 1; l <= 8; ++l) {
      for (int i = 0; i < counts[l - 1]; ++i, ++k) {
        int prefix = code << (8 - l);
        for (int f = 0; f < (1 << (8 - l)); ++f)
          fast[prefix | f] = (vals[k] << 8) | l;
        ++code;
      }
      code <<= 1;
    }
    present = true;
  }
};

struct BitReader {
  // The entropy stream is de-stuffed ONCE up front (FF00 -> FF, restart
  // markers recorded as segment boundaries) so the hot bit-fill is a
  // branch-light 64-bit window refill instead of a per-byte FF check —
  // ~2x on the serving decode stage.
  std::vector<uint8_t> buf;
  std::vector<size_t> seg_start;  // per restart segment
  std::vector<size_t> seg_stop;
  size_t pos = 0, seg_end = 0;
  size_t cur_seg = 0;
  uint64_t bits = 0;  // MSB-aligned
  int nbits = 0;

  BitReader(const uint8_t* p, const uint8_t* end) {
    buf.reserve((size_t)(end - p) + 8);
    seg_start.push_back(0);
    while (p < end) {
      // bulk-copy the run up to the next 0xFF
      const uint8_t* ff =
          (const uint8_t*)memchr(p, 0xFF, (size_t)(end - p));
      if (!ff) {
        buf.insert(buf.end(), p, end);
        break;
      }
      buf.insert(buf.end(), p, ff);
      p = ff;
      uint8_t n = (p + 1 < end) ? p[1] : 0xD9;
      if (n == 0x00) {
        buf.push_back(0xFF);
        p += 2;
      } else if (n >= 0xD0 && n <= 0xD7) {
        seg_stop.push_back(buf.size());
        seg_start.push_back(buf.size());
        p += 2;
      } else {
        break;  // real marker ends the scan
      }
    }
    seg_stop.push_back(buf.size());
    pos = seg_start[0];
    seg_end = seg_stop[0];
    buf.resize(buf.size() + 8, 0x7F);  // pad: unconditional 8B loads
  }

  void fill() {
    if (pos + 8 <= seg_end + 0) {
      uint64_t w;
      std::memcpy(&w, buf.data() + pos, 8);
#if defined(__GNUC__) || defined(__clang__)
      w = __builtin_bswap64(w);
#endif
      const int take = (64 - nbits) >> 3;
      uint64_t chunk = (w >> nbits);
      const int drop = 64 - nbits - take * 8;
      if (drop) chunk &= ~0ULL << drop;
      bits |= chunk;
      nbits += take * 8;
      pos += take;
      return;
    }
    while (nbits <= 56) {
      if (pos >= seg_end) {
        // pad with a non-code pattern; decode stops at block bounds
        bits |= 0x7F7F7F7F7F7F7F7FULL >> nbits;
        nbits = 64;
        return;
      }
      bits |= (uint64_t)buf[pos++] << (56 - nbits);
      nbits += 8;
    }
  }

  int get(int n) {  // n <= 16
    if (nbits < n) fill();
    int v = (int)(bits >> (64 - n));
    bits <<= n;
    nbits -= n;
    return v;
  }

  int peek8() {
    if (nbits < 8) fill();
    return (int)(bits >> 56);
  }

  void skip(int n) {
    bits <<= n;
    nbits -= n;
  }

  void align_restart() {
    bits = 0;
    nbits = 0;
    if (cur_seg + 1 < seg_start.size()) {
      ++cur_seg;
      pos = seg_start[cur_seg];
      seg_end = seg_stop[cur_seg];
    } else {
      pos = seg_end;  // no more segments: everything pads to EOI
    }
  }
};

inline int huff_decode(BitReader& br, const HuffTable& t) {
  int look = br.peek8();
  int32_t f = t.fast[look];
  if (f >= 0) {
    br.skip(f & 0xFF);
    return f >> 8;
  }
  // slow path: walk lengths 9..16
  int code = br.get(8) ;
  for (int l = 9; l <= 16; ++l) {
    code = (code << 1) | br.get(1);
    if (t.maxcode[l] >= 0 && code <= t.maxcode[l] && code >= t.mincode[l])
      return t.vals[t.valptr[l] + (code - t.mincode[l])];
  }
  throw std::runtime_error("bad huffman code");
}

inline int extend(int v, int n) {
  return (n && v < (1 << (n - 1))) ? v - (1 << n) + 1 : v;
}

struct Component {
  int id, h, v, tq;
  int td = 0, ta = 0;
  int bw = 0, bh = 0;  // blocks across the (padded) plane
  int dcpred = 0;
  py::array_t<int16_t> coef;  // [bh*bw*64] natural order
  int16_t* cptr = nullptr;
};

py::dict jpeg_parse(py::bytes data_b) {
  // NOTE: runs WITH the GIL. The callers are the shm-ring decode
  // WORKER PROCESSES (one interpreter each), so releasing it buys
  // nothing; py::array allocation needs it anyway.
  std::string data = data_b;
  py::dict out;
  std::vector<Component> comps;
  {
    const uint8_t* d = (const uint8_t*)data.data();
    size_t n = data.size();
    if (n < 4 || d[0] != 0xFF || d[1] != 0xD8)
      throw std::invalid_argument("not a JPEG (no SOI)");
    size_t pos = 2;
    uint16_t qt[4][64];
    bool qt_set[4] = {};
    HuffTable hdc[4], hac[4];
    int W = 0, H = 0, restart = 0;
    bool sof_seen = false;
    size_t scan_off = 0, scan_end = 0;

    auto rd16 = [&](size_t p) -> int { return (d[p] << 8) | d[p + 1]; };

    while (pos + 4 <= n) {
      if (d[pos] != 0xFF) {
        ++pos;
        continue;
      }
      uint8_t m = d[pos + 1];
      if (m == 0xD8 || m == 0x01 || (m >= 0xD0 && m <= 0xD7)) {
        pos += 2;
        continue;
      }
      if (m == 0xD9) break;  // EOI
      int len = rd16(pos + 2);
      if (len < 2 || pos + 2 + (size_t)len > n)
        throw std::invalid_argument("truncated segment");
      size_t seg = pos + 4;
      if (m == 0xDB) {  // DQT
        size_t p = seg;
        const size_t segend = pos + 2 + len;
        while (p < segend) {
          int pq = d[p] >> 4, tq = d[p] & 15;
          if (tq > 3) throw std::invalid_argument("bad DQT table id");
          ++p;
          if (p + (pq ? 128 : 64) > segend)
            throw std::invalid_argument("truncated DQT");
          for (int i = 0; i < 64; ++i) {
            qt[tq][kZigzag[i]] = pq ? rd16(p + 2 * i) : d[p + i];
          }
          p += pq ? 128 : 64;
          qt_set[tq] = true;
        }
      } else if (m == 0xC4) {  // DHT
        size_t p = seg;
        const size_t segend = pos + 2 + len;
        while (p < segend) {
          int tc = d[p] >> 4, th = d[p] & 15;
          if (tc > 1 || th > 3)
            throw std::invalid_argument("bad DHT table id");
          ++p;
          if (p + 16 > segend)
            throw std::invalid_argument("truncated DHT");
          uint8_t counts[16];
          int nsym = 0;
          for (int i = 0; i < 16; ++i) {
            counts[i] = d[p + i];
            nsym += counts[i];
          }
          p += 16;
          if (nsym > 256 || p + nsym > segend)
            throw std::invalid_argument("bad DHT symbol count");
          (tc ? hac[th] : hdc[th]).build(counts, d + p, nsym);
          p += nsym;
        }
      } else if (m == 0xC0 || m == 0xC1) {  // SOF0/1 baseline
        if (len < 8) throw std::invalid_argument("truncated SOF");
        if (d[seg] != 8) throw std::invalid_argument("not 8-bit");
        H = rd16(seg + 1);
        W = rd16(seg + 3);
        int nc = d[seg + 5];
        if (nc != 1 && nc != 3)
          throw std::invalid_argument("1- or 3-component only");
        if (len < 8 + 3 * nc)
          throw std::invalid_argument("truncated SOF comps");
        if (W <= 0 || H <= 0 || (int64_t)W * H > (64 << 20))
          throw std::invalid_argument("unreasonable dimensions");
        comps.resize(nc);
        for (int c = 0; c < nc; ++c) {
          comps[c].id = d[seg + 6 + 3 * c];
          comps[c].h = d[seg + 7 + 3 * c] >> 4;
          comps[c].v = d[seg + 7 + 3 * c] & 15;
          comps[c].tq = d[seg + 8 + 3 * c];
          if (comps[c].tq > 3)
            throw std::invalid_argument("bad SOF quant table id");
          if (comps[c].h < 1 || comps[c].h > 2 || comps[c].v < 1 ||
              comps[c].v > 2)
            throw std::invalid_argument("unsupported sampling");
        }
        sof_seen = true;
      } else if (m == 0xC2) {
        throw std::invalid_argument("progressive JPEG");
      } else if ((m >= 0xC3 && m <= 0xCF) && m != 0xC4 && m != 0xC8) {
        throw std::invalid_argument("non-baseline SOF");
      } else if (m == 0xDD) {  // DRI
        restart = rd16(seg);
      } else if (m == 0xDA) {  // SOS
        if (!sof_seen) throw std::invalid_argument("SOS before SOF");
        int ns = d[seg];
        if (len < 6 + 2 * ns)
          throw std::invalid_argument("truncated SOS");
        for (int s2 = 0; s2 < ns; ++s2) {
          int cid = d[seg + 1 + 2 * s2];
          int tt = d[seg + 2 + 2 * s2];
          for (auto& c : comps)
            if (c.id == cid) {
              c.td = tt >> 4;
              c.ta = tt & 15;
              if (c.td > 3 || c.ta > 3)
                throw std::invalid_argument("bad SOS huffman table id");
            }
        }
        scan_off = seg + 1 + 2 * ns + 3;
        scan_end = n;
        pos = scan_off;
        break;
      }
      pos += 2 + len;
    }
    if (!scan_off) throw std::invalid_argument("no scan found");

    int hmax = 1, vmax = 1;
    for (auto& c : comps) {
      hmax = std::max(hmax, c.h);
      vmax = std::max(vmax, c.v);
    }
    int mcux = (W + 8 * hmax - 1) / (8 * hmax);
    int mcuy = (H + 8 * vmax - 1) / (8 * vmax);
    for (auto& c : comps) {
      c.bw = mcux * c.h;
      c.bh = mcuy * c.v;
      c.coef = py::array_t<int16_t>((py::ssize_t)c.bh * c.bw * 64);
      c.cptr = (int16_t*)c.coef.request().ptr;
      std::memset(c.cptr, 0, (size_t)c.bh * c.bw * 128);
    }

    BitReader br(d + scan_off, d + scan_end);
    int mcu_count = 0;
    for (int my = 0; my < mcuy; ++my) {
      for (int mx = 0; mx < mcux; ++mx) {
        if (restart && mcu_count && mcu_count % restart == 0) {
          br.align_restart();
          for (auto& c : comps) c.dcpred = 0;
        }
        ++mcu_count;
        for (auto& c : comps) {
          if (!hdc[c.td].present || !hac[c.ta].present)
            throw std::invalid_argument("missing huffman table");
          for (int by = 0; by < c.v; ++by) {
            for (int bx = 0; bx < c.h; ++bx) {
              int16_t* blk =
                  c.cptr +
                  ((size_t)(my * c.v + by) * c.bw + (mx * c.h + bx)) * 64;
              // DC
              int t = huff_decode(br, hdc[c.td]);
              int diff = t ? extend(br.get(t), t) : 0;
              c.dcpred += diff;
              blk[0] = (int16_t)c.dcpred;
              // AC
              int k = 1;
              while (k < 64) {
                int rs = huff_decode(br, hac[c.ta]);
                int r = rs >> 4, s2 = rs & 15;
                if (s2 == 0) {
                  if (r == 15) {
                    k += 16;
                    continue;
                  }
                  break;  // EOB
                }
                k += r;
                if (k > 63) throw std::runtime_error("AC overflow");
                blk[kZigzag[k]] = (int16_t)extend(br.get(s2), s2);
                ++k;
              }
            }
          }
        }
      }
    }

    out["width"] = W;
    out["height"] = H;
    out["ncomp"] = (int)comps.size();
    py::list cl, ql, sl, bl;
    for (auto& c : comps) {
      if (!qt_set[c.tq]) throw std::invalid_argument("missing qtable");
      py::array_t<uint16_t> q(64);
      std::memcpy(q.request().ptr, qt[c.tq], 128);
      ql.append(q);
      cl.append(c.coef);
      sl.append(py::make_tuple(c.h, c.v));
      bl.append(py::make_tuple(c.bh, c.bw));
    }
    out["coef"] = cl;
    out["qt"] = ql;
    out["sampling"] = sl;
    out["blocks"] = bl;
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_jpegc, m) {
  m.doc() = "baseline JPEG entropy decode -> DCT coefficient planes";
  m.def("parse", &jpeg_parse, py::arg("data"));
}
