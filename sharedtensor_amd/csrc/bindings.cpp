// Python bindings for the shared-tensor engine (replaces the reference's Lua
// C-API layer, /root/reference/src/sharedtensor.c:347-477).  The binding is
// torch-header-free: tensors cross the boundary as raw data_ptr() integers,
// so the extension builds with bare hipcc and works for CPU and GPU tensors
// alike.  All blocking entrypoints release the GIL; engine threads never
// touch Python.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "codec_cpu.h"
#include "engine.h"
#include "rccl_transport.h"

namespace py = pybind11;
using namespace shamd;

namespace {

// CPU codec entrypoints for tests (numerics parity vs ops/oracle.py).
py::tuple py_cpu_encode(int codec, uintptr_t delta, int64_t n, double scale_in,
                        uintptr_t payload_out) {
  Codec c = static_cast<Codec>(codec);
  float* d = reinterpret_cast<float*>(delta);
  float scale = scale_in < 0 ? cpu_compute_scale(c, d, n)
                             : static_cast<float>(scale_in);
  cpu_quantize(c, d, n, scale, reinterpret_cast<uint8_t*>(payload_out));
  return py::make_tuple(scale, payload_bytes(c, n));
}

void py_cpu_apply(int codec, uintptr_t payload, int64_t n, double scale,
                  std::vector<uintptr_t> dsts) {
  std::vector<float*> d;
  for (auto p : dsts) d.push_back(reinterpret_cast<float*>(p));
  cpu_apply(static_cast<Codec>(codec), reinterpret_cast<uint8_t*>(payload), n,
            static_cast<float>(scale), d.data(), static_cast<int>(d.size()));
}

double py_cpu_scale(int codec, uintptr_t delta, int64_t n, int stride) {
  return cpu_compute_scale(static_cast<Codec>(codec),
                           reinterpret_cast<float*>(delta), n, stride);
}

// Direct handle on the CDNA4 codec kernels, for numerics tests and ad-hoc
// use (tensors pass as data_ptr integers; caller owns all buffers).
struct DevCodec {
  DevTable tb{};
  Codec c;
  int device;
  bool dbf16;
  void* reduce = nullptr;

  DevCodec(int codec, std::vector<int64_t> sizes, int dev, bool delta_bf16)
      : c(static_cast<Codec>(codec)), device(dev), dbf16(delta_bf16) {
    std::vector<int64_t> offs(sizes.size() + 1), poffs(sizes.size() + 1);
    offs[0] = poffs[0] = 0;
    for (size_t t = 0; t < sizes.size(); ++t) {
      offs[t + 1] = offs[t] + sizes[t];
      poffs[t + 1] = poffs[t] + pad64(sizes[t]);
    }
    if (hipSetDevice(device) != hipSuccess)
      throw std::runtime_error("hipSetDevice failed");
    int T = static_cast<int>(sizes.size());
    if (hipMalloc(&tb.offs, sizeof(int64_t) * (T + 1) * 2) != hipSuccess)
      throw std::runtime_error("hipMalloc failed");
    tb.poffs = tb.offs + (T + 1);
    (void)hipMemcpy(tb.offs, offs.data(), sizeof(int64_t) * (T + 1),
              hipMemcpyHostToDevice);
    (void)hipMemcpy(tb.poffs, poffs.data(), sizeof(int64_t) * (T + 1),
              hipMemcpyHostToDevice);
    tb.T = T;
    tb.n = offs[T];
    tb.pe = poffs[T];
    if (hipMalloc(&reduce, 8 * T) != hipSuccess)
      throw std::runtime_error("hipMalloc failed");
  }
  ~DevCodec() {
    if (tb.offs) (void)hipFree(tb.offs);
    if (reduce) (void)hipFree(reduce);
  }
  void reduce_scales(uintptr_t delta, uintptr_t scales_dev, int stride,
                     uintptr_t stream) {
    hip_reduce_scales(c, reinterpret_cast<const void*>(delta), dbf16, tb,
                      reduce, reinterpret_cast<float*>(scales_dev), stride,
                      reinterpret_cast<hipStream_t>(stream));
  }
  void quantize(uintptr_t delta, uintptr_t scales_dev, uintptr_t payload,
                uintptr_t stream, uintptr_t stats = 0) {
    hip_quantize(c, reinterpret_cast<void*>(delta), dbf16, tb,
                 reinterpret_cast<const float*>(scales_dev),
                 reinterpret_cast<uint8_t*>(payload),
                 reinterpret_cast<hipStream_t>(stream),
                 reinterpret_cast<void*>(stats));
  }
  void finalize_scales(uintptr_t stats, uintptr_t scales_dev, uintptr_t stream) {
    hip_finalize_scales(c, tb, reinterpret_cast<const void*>(stats),
                        reinterpret_cast<float*>(scales_dev), 1,
                        reinterpret_cast<hipStream_t>(stream));
  }
  // dsts[0] = fp32 values (or 0); dsts[1..2] = delta-typed forwards
  void apply(uintptr_t payload, uintptr_t scales_dev,
             std::vector<uintptr_t> dsts, uintptr_t stream) {
    uintptr_t d[3] = {0, 0, 0};
    for (size_t i = 0; i < dsts.size() && i < 3; ++i) d[i] = dsts[i];
    hip_apply(c, reinterpret_cast<const uint8_t*>(payload), tb,
              reinterpret_cast<const float*>(scales_dev),
              reinterpret_cast<float*>(d[0]), reinterpret_cast<void*>(d[1]),
              reinterpret_cast<void*>(d[2]), dbf16,
              reinterpret_cast<hipStream_t>(stream));
  }
};

// dsts[0] = fp32 values (or 0); dsts[1..3] = delta-typed
void py_gpu_add_scatter(uintptr_t src, int64_t n, double alpha,
                        std::vector<uintptr_t> dsts, uintptr_t stream,
                        bool delta_bf16) {
  uintptr_t d[4] = {0, 0, 0, 0};
  for (size_t i = 0; i < dsts.size() && i < 4; ++i) d[i] = dsts[i];
  hip_add_scatter(reinterpret_cast<const float*>(src), n,
                  static_cast<float>(alpha), reinterpret_cast<float*>(d[0]),
                  reinterpret_cast<void*>(d[1]), reinterpret_cast<void*>(d[2]),
                  reinterpret_cast<void*>(d[3]), delta_bf16,
                  reinterpret_cast<hipStream_t>(stream));
}

// dsts[0] = fp32 values (or 0); dsts[1..3] = delta-typed
void py_gpu_fused_sgd(uintptr_t mom, uintptr_t grad, double lr, double mu,
                      int64_t n, std::vector<uintptr_t> dsts, uintptr_t stream,
                      bool delta_bf16) {
  uintptr_t d[4] = {0, 0, 0, 0};
  for (size_t i = 0; i < dsts.size() && i < 4; ++i) d[i] = dsts[i];
  hip_fused_sgd(reinterpret_cast<float*>(mom),
                reinterpret_cast<const float*>(grad), static_cast<float>(lr),
                static_cast<float>(mu), n, reinterpret_cast<float*>(d[0]),
                reinterpret_cast<void*>(d[1]), reinterpret_cast<void*>(d[2]),
                reinterpret_cast<void*>(d[3]), delta_bf16,
                reinterpret_cast<hipStream_t>(stream));
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "MI355X-native shared-tensor engine (CDNA4 HIP + RCCL/TCP)";

  py::class_<Config>(m, "Config")
      .def(py::init<>())
      .def_readwrite("host", &Config::host)
      .def_readwrite("port", &Config::port)
      .def_readwrite("device", &Config::device)
      .def_property(
          "codec", [](const Config& c) { return static_cast<int>(c.codec); },
          [](Config& c, int v) { c.codec = static_cast<Codec>(v); })
      .def_readwrite("snapshot_join", &Config::snapshot_join)
      .def_readwrite("use_rccl", &Config::use_rccl)
      .def_readwrite("reconnect", &Config::reconnect)
      .def_readwrite("preserve_subtree", &Config::preserve_subtree)
      .def_readwrite("keepalive_s", &Config::keepalive_s)
      .def_readwrite("bw_limit", &Config::bw_limit)
      .def_readwrite("min_round_interval_s", &Config::min_round_interval_s)
      .def_readwrite("expected_children", &Config::expected_children)
      .def_readwrite("sizes", &Config::sizes)
      .def_readwrite("explicit_parent", &Config::explicit_parent)
      .def_readwrite("listen_port", &Config::listen_port)
      .def_readwrite("join_timeout_s", &Config::join_timeout_s)
      .def_readwrite("rms_sample_stride", &Config::rms_sample_stride)
      .def_readwrite("lagged_scale", &Config::lagged_scale)
      .def_readwrite("delta_bf16", &Config::delta_bf16)
      .def_readwrite("use_graphs", &Config::use_graphs);

  py::class_<Engine>(m, "Engine")
      .def(py::init<Config>())
      .def("set_values", &Engine::set_values)
      .def("set_link_buffers", &Engine::set_link_buffers)
      .def("start", &Engine::start, py::call_guard<py::gil_scoped_release>())
      .def("add_from", &Engine::add_from,
           py::call_guard<py::gil_scoped_release>())
      .def("copy_to", &Engine::copy_to,
           py::call_guard<py::gil_scoped_release>())
      .def("fused_sgd", &Engine::fused_sgd,
           py::call_guard<py::gil_scoped_release>())
      .def("fused_sgd_bf16", &Engine::fused_sgd_bf16,
           py::call_guard<py::gil_scoped_release>())
      .def("fused_adamw", &Engine::fused_adamw,
           py::call_guard<py::gil_scoped_release>())
      .def("notify_dirty", &Engine::notify_dirty)
      .def("close", &Engine::close, py::call_guard<py::gil_scoped_release>())
      .def("is_master", &Engine::is_master)
      .def("listen_port", &Engine::listen_port)
      .def("last_error", &Engine::last_error)
      .def("reconnect_count", &Engine::reconnect_count)
      .def("recent_scales_sent", &Engine::recent_scales_sent)
      .def("recent_scales_recv", &Engine::recent_scales_recv)
      .def("link_stats", [](Engine& e) {
        py::list out;
        for (auto& s : e.link_stats()) {
          py::dict d;
          d["rounds_sent"] = s.rounds_sent;
          d["rounds_recv"] = s.rounds_recv;
          d["bytes_sent"] = s.bytes_sent;
          d["bytes_recv"] = s.bytes_recv;
          d["last_scale_sent"] = s.last_scale_sent;
          d["last_scale_recv"] = s.last_scale_recv;
          d["active"] = s.active;
          d["dead"] = s.dead;
          d["peer"] = s.peer;
          d["rccl"] = s.rccl;
          out.append(d);
        }
        return out;
      });

  py::class_<DevCodec>(m, "DevCodec")
      .def(py::init<int, std::vector<int64_t>, int, bool>(), py::arg("codec"),
           py::arg("sizes"), py::arg("device"), py::arg("delta_bf16") = false)
      .def("reduce_scales", &DevCodec::reduce_scales)
      .def("quantize", &DevCodec::quantize, py::arg("delta"),
           py::arg("scales"), py::arg("payload"), py::arg("stream"),
           py::arg("stats") = 0)
      .def("finalize_scales", &DevCodec::finalize_scales)
      .def("apply", &DevCodec::apply);
  m.def("gpu_add_scatter", &py_gpu_add_scatter, py::arg("src"), py::arg("n"),
        py::arg("alpha"), py::arg("dsts"), py::arg("stream"),
        py::arg("delta_bf16") = false);
  m.def("gpu_fused_sgd", &py_gpu_fused_sgd, py::arg("mom"), py::arg("grad"),
        py::arg("lr"), py::arg("mu"), py::arg("n"), py::arg("dsts"),
        py::arg("stream"), py::arg("delta_bf16") = false);
  m.def("rccl_loopback_payload", &rccl_loopback_payload,
        "Self ncclSend/ncclRecv of a real payload on one device (1-rank "
        "comm): executes the non-blocking enqueue ordering + stream polling "
        "the 2-GPU xGMI links use and verifies the moved bytes.");
  m.def("rccl_self_test", &rccl_self_test,
        py::call_guard<py::gil_scoped_release>());

  // fused bf16 GELU
  m.def("gelu_fwd", [](uintptr_t x, uintptr_t y, int64_t n, uintptr_t stream) {
    hip_gelu_fwd(reinterpret_cast<const void*>(x), reinterpret_cast<void*>(y),
                 n, reinterpret_cast<hipStream_t>(stream));
  });
  m.def("gelu_bwd", [](uintptr_t dy, uintptr_t x, uintptr_t dx, int64_t n,
                       uintptr_t stream) {
    hip_gelu_bwd(reinterpret_cast<const void*>(dy),
                 reinterpret_cast<const void*>(x),
                 reinterpret_cast<void*>(dx), n,
                 reinterpret_cast<hipStream_t>(stream));
  });

  // fused bf16 cross-entropy
  m.def("ce_fwd", [](uintptr_t logits, uintptr_t targets, uintptr_t loss,
                     uintptr_t row_m, uintptr_t row_lse, int64_t R, int64_t V,
                     uintptr_t stream) {
    hip_ce_fwd(reinterpret_cast<const void*>(logits),
               reinterpret_cast<const int32_t*>(targets),
               reinterpret_cast<float*>(loss), reinterpret_cast<float*>(row_m),
               reinterpret_cast<float*>(row_lse), R, V,
               reinterpret_cast<hipStream_t>(stream));
  });
  m.def("ce_bwd", [](uintptr_t logits, uintptr_t targets, uintptr_t row_lse,
                     uintptr_t dlogits, uintptr_t gscale_dev, double inv_r,
                     int64_t R, int64_t V, uintptr_t stream) {
    hip_ce_bwd(reinterpret_cast<const void*>(logits),
               reinterpret_cast<const int32_t*>(targets),
               reinterpret_cast<const float*>(row_lse),
               reinterpret_cast<void*>(dlogits),
               reinterpret_cast<const float*>(gscale_dev),
               static_cast<float>(inv_r), R, V,
               reinterpret_cast<hipStream_t>(stream));
  });

  // fused bf16 LayerNorm (pointers are bf16 unless named mean/rstd/dgamma/
  // dbeta, which are fp32)
  m.def("ln_fwd", [](uintptr_t x, uintptr_t w, uintptr_t b, uintptr_t y,
                     uintptr_t mean, uintptr_t rstd, int64_t R, int C,
                     double eps, uintptr_t stream) {
    hip_ln_fwd(reinterpret_cast<const void*>(x),
               reinterpret_cast<const void*>(w),
               reinterpret_cast<const void*>(b), reinterpret_cast<void*>(y),
               reinterpret_cast<float*>(mean), reinterpret_cast<float*>(rstd),
               R, C, static_cast<float>(eps),
               reinterpret_cast<hipStream_t>(stream));
  });
  m.def("colsum_bf16", [](uintptr_t x, uintptr_t out, int64_t R, int C,
                          uintptr_t stream) {
    hip_colsum_bf16(reinterpret_cast<const void*>(x),
                    reinterpret_cast<float*>(out), R, C,
                    reinterpret_cast<hipStream_t>(stream));
  });
  m.def("swiglu_fwd", [](uintptr_t x1, uintptr_t x3, uintptr_t y, int64_t n,
                         uintptr_t stream) {
    hip_swiglu_fwd(reinterpret_cast<const void*>(x1),
                   reinterpret_cast<const void*>(x3),
                   reinterpret_cast<void*>(y), n,
                   reinterpret_cast<hipStream_t>(stream));
  });
  m.def("swiglu_bwd", [](uintptr_t dy, uintptr_t x1, uintptr_t x3,
                         uintptr_t dx1, uintptr_t dx3, int64_t n,
                         uintptr_t stream) {
    hip_swiglu_bwd(reinterpret_cast<const void*>(dy),
                   reinterpret_cast<const void*>(x1),
                   reinterpret_cast<const void*>(x3),
                   reinterpret_cast<void*>(dx1),
                   reinterpret_cast<void*>(dx3), n,
                   reinterpret_cast<hipStream_t>(stream));
  });
  m.def("rms_fwd", [](uintptr_t x, uintptr_t w, uintptr_t y, uintptr_t rstd,
                      int64_t R, int C, double eps, uintptr_t stream) {
    hip_rms_fwd(reinterpret_cast<const void*>(x),
                reinterpret_cast<const void*>(w), reinterpret_cast<void*>(y),
                reinterpret_cast<float*>(rstd), R, C, static_cast<float>(eps),
                reinterpret_cast<hipStream_t>(stream));
  });
  m.def("rms_bwd", [](uintptr_t dy, uintptr_t x, uintptr_t w, uintptr_t rstd,
                      uintptr_t dx, uintptr_t dgamma, int64_t R, int C,
                      uintptr_t stream) {
    hip_rms_bwd(reinterpret_cast<const void*>(dy),
                reinterpret_cast<const void*>(x),
                reinterpret_cast<const void*>(w),
                reinterpret_cast<const float*>(rstd),
                reinterpret_cast<void*>(dx), reinterpret_cast<float*>(dgamma),
                R, C, reinterpret_cast<hipStream_t>(stream));
  });
  m.def("ln_bwd", [](uintptr_t dy, uintptr_t x, uintptr_t w, uintptr_t mean,
                     uintptr_t rstd, uintptr_t dx, uintptr_t dgamma,
                     uintptr_t dbeta, int64_t R, int C, uintptr_t stream) {
    hip_ln_bwd(reinterpret_cast<const void*>(dy),
               reinterpret_cast<const void*>(x),
               reinterpret_cast<const void*>(w),
               reinterpret_cast<const float*>(mean),
               reinterpret_cast<const float*>(rstd),
               reinterpret_cast<void*>(dx), reinterpret_cast<float*>(dgamma),
               reinterpret_cast<float*>(dbeta), R, C,
               reinterpret_cast<hipStream_t>(stream));
  });

  m.def("msg_bytes", &Engine::msg_bytes);
  m.def("scales_area", &Engine::scales_area);
  m.def("payload_bytes",
        [](int c, int64_t n) { return payload_bytes(static_cast<Codec>(c), n); });
  m.def("pad64", &pad64);
  m.def("cpu_encode", &py_cpu_encode);
  m.def("cpu_apply", &py_cpu_apply);
  m.def("cpu_scale", &py_cpu_scale);
  m.def("f32_to_e4m3", [](float x) { return static_cast<int>(f32_to_e4m3(x)); });
  m.def("e4m3_to_f32", [](int v) { return e4m3_to_f32(static_cast<uint8_t>(v)); });

#ifdef SHAMD_WITH_HIP
  m.attr("with_hip") = true;
#else
  m.attr("with_hip") = false;
#endif
}
