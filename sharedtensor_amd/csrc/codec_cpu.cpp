#include "codec_cpu.h"

#include <atomic>
#include <cmath>
#include <cstdlib>
#include <cstring>
#include <thread>
#include <vector>

namespace shamd {

// ------------------------------------------------------------ parallel-for
static constexpr int64_t PF_CHUNK = int64_t(1) << 22;  // 4M elems, 64-aligned

static int cpu_codec_threads() {
  static const int t = [] {
    const char* e = std::getenv("SHTENS_CPU_THREADS");
    if (e) {
      int v = std::atoi(e);
      return v < 1 ? 1 : v;
    }
    unsigned hw = std::thread::hardware_concurrency();
    // cap: several link threads may run codecs concurrently and the engine
    // shares the host with training; raise via SHTENS_CPU_THREADS for
    // dedicated 100 GB-scale CPU replicas (BASELINE config 5 rehearsal)
    return static_cast<int>(hw < 1 ? 1 : (hw > 16 ? 16 : hw));
  }();
  return t;
}

void cpu_pfor(int64_t n, const std::function<void(int64_t, int64_t)>& fn) {
  if (n <= 0) return;
  int64_t nchunks = (n + PF_CHUNK - 1) / PF_CHUNK;
  int nt = cpu_codec_threads();
  if (nchunks == 1 || nt == 1) {
    fn(0, n);
    return;
  }
  if (static_cast<int64_t>(nt) > nchunks) nt = static_cast<int>(nchunks);
  std::atomic<int64_t> next{0};
  auto worker = [&] {
    for (;;) {
      int64_t c = next.fetch_add(1, std::memory_order_relaxed);
      if (c >= nchunks) return;
      int64_t lo = c * PF_CHUNK;
      int64_t hi = lo + PF_CHUNK < n ? lo + PF_CHUNK : n;
      fn(lo, hi);
    }
  };
  std::vector<std::thread> ths;
  ths.reserve(nt - 1);
  for (int i = 1; i < nt; ++i) ths.emplace_back(worker);
  worker();
  for (auto& t : ths) t.join();
}

// ----------------------------------------------------------------- fp8 e4m3
uint8_t f32_to_e4m3(float xf) {
  if (xf != xf) return 0x7F;  // NaN
  float cl = xf > 448.f ? 448.f : (xf < -448.f ? -448.f : xf);
  uint32_t u;
  std::memcpy(&u, &cl, 4);
  uint8_t sign = (u >> 24) & 0x80;
  uint32_t abs = u & 0x7FFFFFFFu;
  if (abs == 0) return sign;
  int e = static_cast<int>(abs >> 23) - 127;
  uint32_t mant = (abs & 0x7FFFFFu) | 0x800000u;  // 24-bit with implicit bit
  // normals keep 3 mantissa bits (drop 20); subnormal targets are multiples
  // of 2^-9, needing a deeper shift for smaller exponents
  int shift = (e >= -6) ? 20 : (14 - e);
  if (shift > 24) return sign;  // |x| < 2^-10 half-quantum: rounds to zero
  uint32_t q = mant >> shift;
  uint32_t rem = mant & ((1u << shift) - 1);
  uint32_t half = 1u << (shift - 1);
  if (rem > half || (rem == half && (q & 1))) q++;
  if (e >= -6) {
    if (q == 16) { q = 8; e++; }  // mantissa carry
    return sign | static_cast<uint8_t>(((e + 7) << 3) | (q - 8));
  }
  if (q >= 8) return sign | 0x08;  // rounded up to the smallest normal 2^-6
  return sign | static_cast<uint8_t>(q);
}

float e4m3_to_f32(uint8_t v) {
  int sign = (v & 0x80) ? -1 : 1;
  uint32_t exp = (v >> 3) & 0xF;
  uint32_t man = v & 0x7;
  if (exp == 0xF && man == 0x7) return NAN;  // e4m3fn NaN
  if (exp == 0) return sign * std::ldexp(static_cast<float>(man), -9);
  return sign * std::ldexp(8.0f + man, static_cast<int>(exp) - 10);
}

// ------------------------------------------------------------ pow2 helpers
float pow2_floor_f(double x) {
  if (!(x > 0.0) || std::isinf(x) || std::isnan(x)) return 0.0f;
  int e;
  std::frexp(x, &e);  // x = m * 2^e, m in [0.5, 1)
  return static_cast<float>(std::ldexp(1.0, e - 1));
}

float pow2_ceil_f(double x) {
  if (!(x > 0.0) || std::isinf(x) || std::isnan(x)) return 0.0f;
  int e;
  double m = std::frexp(x, &e);
  return static_cast<float>(std::ldexp(1.0, m == 0.5 ? e - 1 : e));
}

// ------------------------------------------------------------------ scales
float cpu_compute_scale(Codec c, const float* delta, int64_t n, int stride) {
  if (n <= 0) return 0.0f;
  if (stride < 1) stride = 1;
  // per-chunk partials combined IN CHUNK ORDER: the result is bit-identical
  // regardless of worker count (chunk decomposition depends only on n)
  int64_t nchunks = (n + PF_CHUNK - 1) / PF_CHUNK;
  if (c == Codec::OneBit) {
    std::vector<double> part(static_cast<size_t>(nchunks), 0.0);
    std::vector<int64_t> cnts(static_cast<size_t>(nchunks), 0);
    cpu_pfor(n, [&](int64_t lo, int64_t hi) {
      int64_t ci = lo / PF_CHUNK;
      // strided sampling restarts on the chunk-local grid; with stride=1
      // (the default) this is the exact sum of squares
      double ss = 0.0;
      int64_t cnt = 0;
      for (int64_t i = lo; i < hi; i += stride, ++cnt) {
        double d = atomic_load_f32(delta + i);
        ss += d * d;
      }
      part[static_cast<size_t>(ci)] = ss;
      cnts[static_cast<size_t>(ci)] = cnt;
    });
    double ss = 0.0;
    int64_t cnt = 0;
    for (int64_t ci = 0; ci < nchunks; ++ci) {
      ss += part[static_cast<size_t>(ci)];
      cnt += cnts[static_cast<size_t>(ci)];
    }
    return pow2_floor_f(std::sqrt(ss / static_cast<double>(cnt)));
  }
  std::vector<float> partm(static_cast<size_t>(nchunks), 0.0f);
  cpu_pfor(n, [&](int64_t lo, int64_t hi) {
    int64_t ci = lo / PF_CHUNK;
    float mx = 0.0f;
    for (int64_t i = lo; i < hi; i += stride) {
      float a = std::fabs(atomic_load_f32(delta + i));
      if (a > mx) mx = a;
    }
    partm[static_cast<size_t>(ci)] = mx;
  });
  float mx = 0.0f;
  for (float m : partm)
    if (m > mx) mx = m;
  if (mx == 0.0f || std::isnan(mx) || std::isinf(mx)) return 0.0f;
  return pow2_ceil_f(mx / (c == Codec::Fp8 ? 448.0 : 7.0));
}

// ---------------------------------------------------------------- quantize
void cpu_quantize(Codec c, float* delta, int64_t n, float scale, uint8_t* payload) {
  int64_t pb = payload_bytes(c, n);
  std::memset(payload, 0, pb);
  if (scale == 0.0f) return;
  // parallel over 64-aligned chunks: no packed-payload byte straddles a
  // chunk, so the |= writes below are worker-exclusive
  switch (c) {
    case Codec::OneBit:
      // residual > 0 -> bit 0, send +scale; else bit 1, send -scale
      // (sharedtensor.c:166-174); debit with an atomic so adds that land
      // between the read and the update are never lost.
      cpu_pfor(n, [&](int64_t lo, int64_t hi) {
        for (int64_t i = lo; i < hi; ++i) {
          float v = atomic_load_f32(delta + i);
          float sent;
          if (v > 0.0f) {
            sent = scale;
          } else {
            payload[i / 8] |= 1u << (i % 8);
            sent = -scale;
          }
          atomic_add_f32(delta + i, -sent);
        }
      });
      break;
    case Codec::Fp8: {
      float inv = 1.0f / scale;
      cpu_pfor(n, [&](int64_t lo, int64_t hi) {
        for (int64_t i = lo; i < hi; ++i) {
          float v = atomic_load_f32(delta + i);
          float sc = v * inv;
          sc = sc > 448.f ? 448.f : (sc < -448.f ? -448.f : sc);
          uint8_t q = f32_to_e4m3(sc);
          payload[i] = q;
          atomic_add_f32(delta + i, -e4m3_to_f32(q) * scale);
        }
      });
      break;
    }
    case Codec::Int4: {
      float inv = 1.0f / scale;
      cpu_pfor(n, [&](int64_t lo, int64_t hi) {
        for (int64_t i = lo; i < hi; ++i) {
          float v = atomic_load_f32(delta + i);
          float r = std::nearbyintf(v * inv);  // round-to-nearest-even
          r = r > 7.f ? 7.f : (r < -7.f ? -7.f : r);
          int8_t q = static_cast<int8_t>(r);
          payload[i / 2] |= static_cast<uint8_t>(q & 0xF) << ((i % 2) * 4);
          atomic_add_f32(delta + i, -static_cast<float>(q) * scale);
        }
      });
      break;
    }
  }
}

// ------------------------------------------------------------------- apply
void cpu_apply(Codec c, const uint8_t* payload, int64_t n, float scale,
               float* const* dsts, int ndst) {
  if (scale == 0.0f) return;
  cpu_pfor(n, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) {
      float v = 0.0f;
      switch (c) {
        case Codec::OneBit: {
          int bit = (payload[i / 8] >> (i % 8)) & 1;
          v = bit ? -scale : scale;  // save_deltas, sharedtensor.c:106-111
          break;
        }
        case Codec::Fp8:
          v = e4m3_to_f32(payload[i]) * scale;
          break;
        case Codec::Int4: {
          int8_t q =
              static_cast<int8_t>((payload[i / 2] >> ((i % 2) * 4)) & 0xF);
          if (q > 7) q -= 16;
          v = static_cast<float>(q) * scale;
          break;
        }
      }
      if (v != 0.0f)
        for (int d = 0; d < ndst; ++d) atomic_add_f32(dsts[d] + i, v);
    }
  });
}

void cpu_add_scatter(const float* src, int64_t n, float* const* dsts, int ndst) {
  cpu_pfor(n, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) {
      float v = src[i];
      if (v != 0.0f)
        for (int d = 0; d < ndst; ++d) atomic_add_f32(dsts[d] + i, v);
    }
  });
}

}  // namespace shamd
