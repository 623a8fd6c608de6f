// Image data pipeline: RecordIO reader + own baseline JPEG decoder +
// threaded decode/augment into prefetched batches.
//
// Reference parity: src/io/iter_image_recordio_2.cc:715-887
// (ImageRecordIter: dmlc ThreadedIter chunks -> OMP decode+augment into
// batch TBlobs -> prefetcher) and src/io/image_aug_default.cc.  The
// reference decodes with OpenCV/libjpeg-turbo; this container ships no
// JPEG dev headers, so the decoder below implements baseline sequential
// JFIF itself (DQT/SOF0/DHT/SOS, huffman, dequant, AAN-style IDCT,
// 4:4:4/4:2:2/4:2:0 upsampling, restart markers).  Progressive JPEG is
// rejected with a clear error.
//
// Augment (reference defaults): resize shorter side, random/center crop,
// horizontal mirror, fp32 scale or raw uint8 output, NHWC.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <atomic>
#include <cmath>
#include <cstring>
#include <fstream>
#include <cstdlib>
#include <random>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

// ===========================================================================
// baseline JPEG decoder
// ===========================================================================
struct HuffTable {
  // canonical tables: mincode/maxcode/valptr per length
  int32_t mincode[17] = {0};
  int32_t maxcode[17] = {0};
  int32_t valptr[17] = {0};
  std::vector<uint8_t> vals;
  bool defined = false;

  void Build(const uint8_t counts[16], const uint8_t* symbols, int nsym) {
    vals.assign(symbols, symbols + nsym);
    int code = 0, k = 0;
    for (int l = 1; l <= 16; ++l) {
      valptr[l] = k;
      mincode[l] = code;
      code += counts[l - 1];
      k += counts[l - 1];
      maxcode[l] = code - 1;
      if (counts[l - 1] == 0) maxcode[l] = -1;
      code <<= 1;
    }
    defined = true;
  }
};

struct BitReader {
  const uint8_t* p;
  const uint8_t* end;
  uint32_t bits = 0;
  int nbits = 0;
  bool hit_marker = false;

  void Refill() {
    while (nbits <= 24) {
      if (p >= end) {
        bits |= 0 /* pad zeros past end */;
        nbits += 8;
        continue;
      }
      uint8_t b = *p;
      if (b == 0xFF) {
        uint8_t nx = (p + 1 < end) ? p[1] : 0xD9;
        if (nx == 0x00) {
          p += 2;  // stuffed byte
        } else {
          // real marker: stop consuming (caller handles RST/EOI)
          hit_marker = true;
          bits |= 0;
          nbits += 8;
          continue;
        }
      } else {
        ++p;
      }
      bits |= (uint32_t)b << (24 - nbits);
      nbits += 8;
    }
  }

  int GetBit() {
    if (nbits < 1) Refill();
    int v = (bits >> 31) & 1;
    bits <<= 1;
    --nbits;
    return v;
  }

  int GetBits(int n) {
    int v = 0;
    for (int i = 0; i < n; ++i) v = (v << 1) | GetBit();
    return v;
  }

  void AlignAndSkipRst() {
    // byte-align and consume a restart marker if present
    bits = 0;
    nbits = 0;
    hit_marker = false;
    while (p + 1 < end && p[0] == 0xFF && p[1] >= 0xD0 && p[1] <= 0xD7)
      p += 2;
  }
};

inline int HuffDecode(BitReader* br, const HuffTable& t) {
  int code = br->GetBit();
  for (int l = 1; l <= 16; ++l) {
    if (t.maxcode[l] >= 0 && code <= t.maxcode[l])
      return t.vals[t.valptr[l] + code - t.mincode[l]];
    code = (code << 1) | br->GetBit();
  }
  throw std::runtime_error("jpeg: invalid huffman code");
}

inline int Extend(int v, int n) {
  return (n && v < (1 << (n - 1))) ? v - (1 << n) + 1 : v;
}

const uint8_t kZigzag[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

// simple float separable IDCT (clarity > speed; the decode threads give
// the pipeline its throughput)
void Idct8x8(const float* in, float* out) {
  static float c[8][8];
  static bool init = [] {
    for (int x = 0; x < 8; ++x)
      for (int u = 0; u < 8; ++u)
        c[x][u] = (u == 0 ? 0.353553390593f : 0.5f) *
                  cosf((2 * x + 1) * u * 3.14159265358979f / 16.f);
    return true;
  }();
  (void)init;
  float tmp[64];
  for (int y = 0; y < 8; ++y)
    for (int x = 0; x < 8; ++x) {
      float s = 0;
      for (int u = 0; u < 8; ++u) s += c[x][u] * in[y * 8 + u];
      tmp[y * 8 + x] = s;
    }
  for (int x = 0; x < 8; ++x)
    for (int y = 0; y < 8; ++y) {
      float s = 0;
      for (int v = 0; v < 8; ++v) s += c[y][v] * tmp[v * 8 + x];
      out[y * 8 + x] = s;
    }
}

struct JpegImage {
  int w = 0, h = 0, c = 0;
  std::vector<uint8_t> rgb;  // HWC, c==3 (gray replicated)
};

struct Component {
  int id, hs, vs, tq;
  int td = 0, ta = 0;
  int dc_pred = 0;
  int bw = 0, bh = 0;  // blocks per MCU row/col overall
  std::vector<float> plane;
  int pw = 0, ph = 0;
};

JpegImage DecodeJpeg(const uint8_t* d, size_t n) {
  if (n < 4 || d[0] != 0xFF || d[1] != 0xD8)
    throw std::runtime_error("jpeg: missing SOI");
  size_t pos = 2;
  uint16_t qt[4][64] = {};
  HuffTable hdc[4], hac[4];
  Component comp[4];
  int ncomp = 0, W = 0, H = 0, restart = 0;
  int hmax = 1, vmax = 1;

  auto rd16 = [&](size_t at) { return (d[at] << 8) | d[at + 1]; };

  while (pos + 4 <= n) {
    if (d[pos] != 0xFF) throw std::runtime_error("jpeg: marker desync");
    uint8_t m = d[pos + 1];
    pos += 2;
    if (m == 0xD9) break;  // EOI
    if (m == 0x01 || (m >= 0xD0 && m <= 0xD7)) continue;
    size_t len = rd16(pos);
    size_t seg = pos + 2, seg_end = pos + len;
    pos += len;
    switch (m) {
      case 0xDB:  // DQT
        while (seg < seg_end) {
          int pq = d[seg] >> 4, tq_ = d[seg] & 15;
          ++seg;
          for (int i = 0; i < 64; ++i) {
            qt[tq_][i] = pq ? rd16(seg) : d[seg];
            seg += pq ? 2 : 1;
          }
        }
        break;
      case 0xC0:
      case 0xC1: {  // SOF0/1 baseline
        H = rd16(seg + 1);
        W = rd16(seg + 3);
        ncomp = d[seg + 5];
        if (ncomp != 1 && ncomp != 3)
          throw std::runtime_error("jpeg: unsupported component count");
        for (int i = 0; i < ncomp; ++i) {
          comp[i].id = d[seg + 6 + 3 * i];
          comp[i].hs = d[seg + 7 + 3 * i] >> 4;
          comp[i].vs = d[seg + 7 + 3 * i] & 15;
          comp[i].tq = d[seg + 8 + 3 * i];
          hmax = std::max(hmax, comp[i].hs);
          vmax = std::max(vmax, comp[i].vs);
        }
        break;
      }
      case 0xC2:
        throw std::runtime_error(
            "jpeg: progressive JPEG not supported (baseline decoder)");
      case 0xC4:  // DHT
        while (seg < seg_end) {
          int tc = d[seg] >> 4, th = d[seg] & 15;
          ++seg;
          uint8_t counts[16];
          int nsym = 0;
          for (int i = 0; i < 16; ++i) {
            counts[i] = d[seg + i];
            nsym += counts[i];
          }
          (tc ? hac[th] : hdc[th]).Build(counts, d + seg + 16, nsym);
          seg += 16 + nsym;
        }
        break;
      case 0xDD:  // DRI
        restart = rd16(seg);
        break;
      case 0xDA: {  // SOS
        int ns = d[seg];
        for (int i = 0; i < ns; ++i) {
          int cid = d[seg + 1 + 2 * i];
          for (int j = 0; j < ncomp; ++j)
            if (comp[j].id == cid) {
              comp[j].td = d[seg + 2 + 2 * i] >> 4;
              comp[j].ta = d[seg + 2 + 2 * i] & 15;
            }
        }
        // entropy-coded data starts at seg_end
        int mcux = (W + 8 * hmax - 1) / (8 * hmax);
        int mcuy = (H + 8 * vmax - 1) / (8 * vmax);
        for (int i = 0; i < ncomp; ++i) {
          comp[i].pw = mcux * 8 * comp[i].hs;
          comp[i].ph = mcuy * 8 * comp[i].vs;
          comp[i].plane.assign((size_t)comp[i].pw * comp[i].ph, 0.f);
          comp[i].dc_pred = 0;
        }
        BitReader br{d + seg_end, d + n};
        float blk[64], px[64];
        int mcu_count = 0;
        for (int my = 0; my < mcuy; ++my)
          for (int mx = 0; mx < mcux; ++mx) {
            if (restart && mcu_count && mcu_count % restart == 0) {
              br.AlignAndSkipRst();
              for (int i = 0; i < ncomp; ++i) comp[i].dc_pred = 0;
            }
            ++mcu_count;
            for (int i = 0; i < ncomp; ++i)
              for (int by = 0; by < comp[i].vs; ++by)
                for (int bx = 0; bx < comp[i].hs; ++bx) {
                  std::memset(blk, 0, sizeof(blk));
                  int t = HuffDecode(&br, hdc[comp[i].td]);
                  int diff = Extend(br.GetBits(t), t);
                  comp[i].dc_pred += diff;
                  blk[0] = (float)comp[i].dc_pred * qt[comp[i].tq][0];
                  for (int k = 1; k < 64;) {
                    int rs = HuffDecode(&br, hac[comp[i].ta]);
                    int r = rs >> 4, s = rs & 15;
                    if (s == 0) {
                      if (r == 15) {
                        k += 16;
                        continue;
                      }
                      break;  // EOB
                    }
                    k += r;
                    if (k > 63)
                      throw std::runtime_error("jpeg: AC index overflow");
                    int v = Extend(br.GetBits(s), s);
                    blk[kZigzag[k]] = (float)v * qt[comp[i].tq][k];
                    ++k;
                  }
                  Idct8x8(blk, px);
                  int ox = (mx * comp[i].hs + bx) * 8;
                  int oy = (my * comp[i].vs + by) * 8;
                  for (int y = 0; y < 8; ++y)
                    for (int x = 0; x < 8; ++x)
                      comp[i].plane[(size_t)(oy + y) * comp[i].pw + ox + x] =
                          px[y * 8 + x] + 128.f;
                }
          }
        // color convert
        JpegImage img;
        img.w = W;
        img.h = H;
        img.c = 3;
        img.rgb.resize((size_t)W * H * 3);
        for (int y = 0; y < H; ++y)
          for (int x = 0; x < W; ++x) {
            float Y = comp[0].plane[(size_t)(y * comp[0].vs / vmax) *
                                        comp[0].pw +
                                    x * comp[0].hs / hmax];
            float R, G, B;
            if (ncomp == 3) {
              float cb = comp[1].plane[(size_t)(y * comp[1].vs / vmax) *
                                           comp[1].pw +
                                       x * comp[1].hs / hmax] -
                         128.f;
              float cr = comp[2].plane[(size_t)(y * comp[2].vs / vmax) *
                                           comp[2].pw +
                                       x * comp[2].hs / hmax] -
                         128.f;
              R = Y + 1.402f * cr;
              G = Y - 0.344136f * cb - 0.714136f * cr;
              B = Y + 1.772f * cb;
            } else {
              R = G = B = Y;
            }
            auto clamp = [](float v) {
              return (uint8_t)(v < 0 ? 0 : (v > 255 ? 255 : v + 0.5f));
            };
            size_t o = ((size_t)y * W + x) * 3;
            img.rgb[o] = clamp(R);
            img.rgb[o + 1] = clamp(G);
            img.rgb[o + 2] = clamp(B);
          }
        return img;
      }
      default:
        break;  // APPn/COM/etc: skip
    }
  }
  throw std::runtime_error("jpeg: no SOS segment");
}

// bilinear resize (uint8 HWC3)
void Resize(const uint8_t* src, int sh, int sw, uint8_t* dst, int dh,
            int dw) {
  for (int y = 0; y < dh; ++y) {
    float fy = (y + 0.5f) * sh / dh - 0.5f;
    int y0 = (int)floorf(fy);
    float wy = fy - y0;
    int y1 = std::min(y0 + 1, sh - 1);
    y0 = std::max(y0, 0);
    for (int x = 0; x < dw; ++x) {
      float fx = (x + 0.5f) * sw / dw - 0.5f;
      int x0 = (int)floorf(fx);
      float wx = fx - x0;
      int x1 = std::min(x0 + 1, sw - 1);
      x0 = std::max(x0, 0);
      for (int ch = 0; ch < 3; ++ch) {
        float v00 = src[((size_t)y0 * sw + x0) * 3 + ch];
        float v01 = src[((size_t)y0 * sw + x1) * 3 + ch];
        float v10 = src[((size_t)y1 * sw + x0) * 3 + ch];
        float v11 = src[((size_t)y1 * sw + x1) * 3 + ch];
        float v = v00 * (1 - wy) * (1 - wx) + v01 * (1 - wy) * wx +
                  v10 * wy * (1 - wx) + v11 * wy * wx;
        dst[((size_t)y * dw + x) * 3 + ch] = (uint8_t)(v + 0.5f);
      }
    }
  }
}

// ===========================================================================
// RecordIO + threaded batcher
// ===========================================================================
constexpr uint32_t kRecMagic = 0xced7230a;

#pragma pack(push, 1)
struct IRHeader {
  uint32_t flag;
  float label;
  uint64_t id;
  uint64_t id2;
};
#pragma pack(pop)

class ImageRecordIter {
 public:
  ImageRecordIter(const std::string& path, int batch, int out_h, int out_w,
                  int threads, bool shuffle, bool rand_crop,
                  bool rand_mirror, int resize_shorter, uint64_t seed)
      : batch_(batch), oh_(out_h), ow_(out_w),
        threads_(threads > 0 ? threads
                             : (int)std::thread::hardware_concurrency()),
        shuffle_(shuffle), rand_crop_(rand_crop), rand_mirror_(rand_mirror),
        resize_(resize_shorter), rng_(seed) {
    std::ifstream f(path, std::ios::binary | std::ios::ate);
    if (!f) throw std::runtime_error("cannot open " + path);
    size_t sz = f.tellg();
    data_.resize(sz);
    f.seekg(0);
    f.read((char*)data_.data(), sz);
    // index the records
    size_t pos = 0;
    while (pos + 8 <= sz) {
      uint32_t magic, lrec;
      std::memcpy(&magic, data_.data() + pos, 4);
      std::memcpy(&lrec, data_.data() + pos + 4, 4);
      if (magic != kRecMagic)
        throw std::runtime_error("bad recordio magic in " + path);
      uint32_t len = lrec & ((1u << 29) - 1);
      offsets_.push_back({pos + 8, len});
      pos += 8 + ((len + 3) & ~3u);
    }
    Reset();
  }

  size_t size() const { return offsets_.size(); }

  void Reset() {
    order_.resize(offsets_.size());
    for (size_t i = 0; i < order_.size(); ++i) order_[i] = i;
    if (shuffle_) std::shuffle(order_.begin(), order_.end(), rng_);
    cursor_ = 0;
  }

  // fill the caller's uint8 NHWC batch + fp32 labels; returns #valid rows
  int NextInto(uint8_t* out, float* labels) {
    if (cursor_ >= order_.size()) return 0;
    size_t begin = cursor_;
    size_t take = std::min((size_t)batch_, order_.size() - begin);
    cursor_ += take;
    std::atomic<size_t> next{0};
    std::vector<std::thread> pool;
    std::vector<std::string> errors(threads_);
    for (int t = 0; t < threads_; ++t) {
      pool.emplace_back([&, t] {
        // per-thread rng seeded off the batch position: reproducible
        std::mt19937_64 trng(rng_seed_base_ + begin * 1315423911u + t);
        for (;;) {
          size_t i = next.fetch_add(1);
          if (i >= take) return;
          try {
            DecodeOne(order_[begin + i],
                      out + (size_t)i * oh_ * ow_ * 3, labels + i, &trng);
          } catch (const std::exception& e) {
            errors[t] = e.what();
            return;
          }
        }
      });
    }
    for (auto& th : pool) th.join();
    for (auto& e : errors)
      if (!e.empty()) throw std::runtime_error(e);
    return (int)take;
  }

 private:
  void DecodeOne(size_t rec, uint8_t* dst, float* label,
                 std::mt19937_64* trng) {
    auto [off, len] = offsets_[rec];
    const uint8_t* p = data_.data() + off;
    if (len < sizeof(IRHeader)) throw std::runtime_error("short record");
    IRHeader hdr;
    std::memcpy(&hdr, p, sizeof(hdr));
    *label = hdr.label;
    const uint8_t* payload = p + sizeof(IRHeader);
    size_t pl = len - sizeof(IRHeader);
    // flag>0: extra label floats precede the image (reference IRHeader)
    payload += hdr.flag * 4;
    pl -= hdr.flag * 4;

    JpegImage img;
    if (pl >= 2 && payload[0] == 0xFF && payload[1] == 0xD8) {
      img = DecodeJpeg(payload, pl);
    } else if (pl >= 8) {
      // raw mode: u32 h, u32 w, then h*w*3 uint8 (im2rec --raw)
      uint32_t rh, rw;
      std::memcpy(&rh, payload, 4);
      std::memcpy(&rw, payload + 4, 4);
      if (8 + (size_t)rh * rw * 3 > pl)
        throw std::runtime_error("raw record truncated");
      img.h = rh;
      img.w = rw;
      img.c = 3;
      img.rgb.assign(payload + 8, payload + 8 + (size_t)rh * rw * 3);
    } else {
      throw std::runtime_error("unrecognized image payload");
    }

    // resize shorter side
    std::vector<uint8_t> resized;
    const uint8_t* src = img.rgb.data();
    int sh = img.h, sw = img.w;
    int target = resize_ > 0 ? resize_ : 0;
    if (target > 0 && std::min(sh, sw) != target) {
      int nh, nw;
      if (sh < sw) {
        nh = target;
        nw = (int)((int64_t)sw * target / sh);
      } else {
        nw = target;
        nh = (int)((int64_t)sh * target / sw);
      }
      resized.resize((size_t)nh * nw * 3);
      Resize(src, sh, sw, resized.data(), nh, nw);
      src = resized.data();
      sh = nh;
      sw = nw;
    }
    if (sh < oh_ || sw < ow_) {
      // upscale to fit the crop
      int nh = std::max(sh, oh_), nw = std::max(sw, ow_);
      std::vector<uint8_t> up((size_t)nh * nw * 3);
      Resize(src, sh, sw, up.data(), nh, nw);
      resized = std::move(up);
      src = resized.data();
      sh = nh;
      sw = nw;
    }
    // crop
    int y0, x0;
    if (rand_crop_) {
      y0 = (int)((*trng)() % (uint64_t)(sh - oh_ + 1));
      x0 = (int)((*trng)() % (uint64_t)(sw - ow_ + 1));
    } else {
      y0 = (sh - oh_) / 2;
      x0 = (sw - ow_) / 2;
    }
    bool mirror = rand_mirror_ && ((*trng)() & 1);
    for (int y = 0; y < oh_; ++y) {
      const uint8_t* row = src + ((size_t)(y0 + y) * sw + x0) * 3;
      uint8_t* drow = dst + (size_t)y * ow_ * 3;
      if (!mirror) {
        std::memcpy(drow, row, (size_t)ow_ * 3);
      } else {
        for (int x = 0; x < ow_; ++x) {
          drow[x * 3] = row[(ow_ - 1 - x) * 3];
          drow[x * 3 + 1] = row[(ow_ - 1 - x) * 3 + 1];
          drow[x * 3 + 2] = row[(ow_ - 1 - x) * 3 + 2];
        }
      }
    }
  }

  int batch_, oh_, ow_, threads_;
  bool shuffle_, rand_crop_, rand_mirror_;
  int resize_;
  std::mt19937_64 rng_;
  uint64_t rng_seed_base_ = 0x9E3779B97F4A7C15ull;
  std::vector<uint8_t> data_;
  std::vector<std::pair<size_t, uint32_t>> offsets_;
  std::vector<size_t> order_;
  size_t cursor_ = 0;
};

// ---------------------------------------------------------------------------
// CSVIter (reference src/io/iter_csv.cc): C++ CSV parse into float
// row-batches; optional separate label file, round-robin batching.
// ---------------------------------------------------------------------------
class CsvIter {
 public:
  CsvIter(const std::string& data_path, const std::string& label_path,
          int batch, int row_width, int label_width)
      : batch_(batch), dwidth_(row_width), lwidth_(label_width) {
    ParseFile(data_path, dwidth_, &data_);
    if (!label_path.empty()) {
      ParseFile(label_path, lwidth_, &labels_);
      if (data_.size() / dwidth_ != labels_.size() / lwidth_)
        throw std::runtime_error("CSVIter: data/label row count mismatch");
    }
    nrows_ = data_.size() / dwidth_;
  }

  size_t size() const { return nrows_; }
  void Reset() { cursor_ = 0; }

  // fills caller buffers; returns rows delivered (< batch at EOF)
  int NextInto(float* data_out, float* label_out) {
    int got = 0;
    while (got < batch_ && cursor_ < nrows_) {
      std::memcpy(data_out + (size_t)got * dwidth_,
                  data_.data() + cursor_ * dwidth_, dwidth_ * 4);
      if (label_out) {
        if (!labels_.empty())
          std::memcpy(label_out + (size_t)got * lwidth_,
                      labels_.data() + cursor_ * lwidth_, lwidth_ * 4);
        else
          label_out[(size_t)got * lwidth_] = 0.f;
      }
      ++cursor_;
      ++got;
    }
    return got;
  }

 private:
  static void ParseFile(const std::string& path, int width,
                        std::vector<float>* out) {
    std::ifstream f(path);
    if (!f.good())
      throw std::runtime_error("CSVIter: cannot open " + path);
    std::string line;
    while (std::getline(f, line)) {
      if (line.empty()) continue;
      const char* p = line.c_str();
      int n = 0;
      while (*p && n < width) {
        char* end = nullptr;
        float v = std::strtof(p, &end);
        if (end == p) break;
        out->push_back(v);
        ++n;
        p = end;
        while (*p == ',' || *p == ' ' || *p == '\t') ++p;
      }
      if (n != width)
        throw std::runtime_error("CSVIter: row has " + std::to_string(n) +
                                 " fields, expected " +
                                 std::to_string(width));
    }
  }

  int batch_, dwidth_, lwidth_;
  size_t nrows_ = 0, cursor_ = 0;
  std::vector<float> data_, labels_;
};

}  // namespace

PYBIND11_MODULE(_imageio, m) {
  m.doc() = "RecordIO image pipeline: own baseline JPEG decoder + "
            "threaded decode/augment (reference ImageRecordIter)";

  m.def("decode_jpeg", [](py::bytes data) {
    std::string s = data;
    JpegImage img = DecodeJpeg((const uint8_t*)s.data(), s.size());
    std::vector<py::ssize_t> shp = {img.h, img.w, 3};
    py::array_t<uint8_t> out(shp);
    std::memcpy(out.mutable_data(), img.rgb.data(), img.rgb.size());
    return out;
  });

  py::class_<ImageRecordIter>(m, "ImageRecordIter")
      .def(py::init<const std::string&, int, int, int, int, bool, bool,
                    bool, int, uint64_t>(),
           py::arg("path"), py::arg("batch_size"), py::arg("out_h"),
           py::arg("out_w"), py::arg("threads") = 0,
           py::arg("shuffle") = false, py::arg("rand_crop") = false,
           py::arg("rand_mirror") = false, py::arg("resize_shorter") = 0,
           py::arg("seed") = 0)
      .def_property_readonly("size", &ImageRecordIter::size)
      .def("reset", &ImageRecordIter::Reset)
      .def("next_into",
           [](ImageRecordIter& it, uintptr_t data_ptr, uintptr_t label_ptr) {
             // caller-provided buffers (e.g. pinned host memory from the
             // native runtime) — zero extra host copies
             int got;
             {
               py::gil_scoped_release rel;
               got = it.NextInto((uint8_t*)data_ptr, (float*)label_ptr);
             }
             return got;
           })
      .def("next_batch", [](ImageRecordIter& it, int batch, int oh, int ow) {
        std::vector<py::ssize_t> shp = {batch, oh, ow, 3};
        py::array_t<uint8_t> data(shp);
        py::array_t<float> labels(batch);
        int got;
        {
          py::gil_scoped_release rel;
          got = it.NextInto(data.mutable_data(), labels.mutable_data());
        }
        return py::make_tuple(got, data, labels);
      });

  py::class_<CsvIter>(m, "CsvIter")
      .def(py::init<const std::string&, const std::string&, int, int, int>(),
           py::arg("data_path"), py::arg("label_path") = "",
           py::arg("batch_size") = 1, py::arg("row_width") = 1,
           py::arg("label_width") = 1)
      .def_property_readonly("size", &CsvIter::size)
      .def("reset", &CsvIter::Reset)
      .def("next_batch", [](CsvIter& it, int batch, int dwidth, int lwidth) {
        py::array_t<float> data({(py::ssize_t)batch, (py::ssize_t)dwidth});
        py::array_t<float> labels({(py::ssize_t)batch,
                                   (py::ssize_t)lwidth});
        int got = it.NextInto(data.mutable_data(), labels.mutable_data());
        return py::make_tuple(got, data, labels);
      });
}
