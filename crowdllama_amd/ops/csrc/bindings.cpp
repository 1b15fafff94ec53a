// pybind11 module `crowdllama_amd.ops._core` — Python face of the
// MI355X-native engine.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <rccl/rccl.h>

#include "engine.h"
#include "../csrc/tokenizer_api.h"

namespace py = pybind11;
using namespace cla;

namespace cla {
void launch_gemv_test(const void*, const void*, const float*, const float*,
                      float*, int, int, int, int, int, size_t, size_t);
void launch_mfma_probe_test(const uint16_t*, const uint16_t*, float*);
double bench_gemv(const void*, const void*, int, int, int, int, int, size_t,
                  size_t, int);
void launch_gemm_test(const void*, const void*, const float*, float*, int,
                      int, int, int, size_t, size_t);
double bench_gemm(int, int, int, int, int);
double bench_gemv_g(int, int, int, int, int);
double bench_membw(int nt, int mb, int wgs, int iters);
void launch_gemv_q8_test(const void*, const void*, const float*, float*, int,
                         int, int, int, size_t, size_t);
double bench_gemv_q8(int, int, int, int, int);
std::vector<uint8_t> slice_cols_test(int32_t, const uint8_t*, int64_t,
                                     int64_t, int64_t, int64_t);
std::vector<float> test_rccl_graph_1rank(const std::vector<float>&);
void launch_gemm_i8_test(const void*, const void*, const float*, float*,
                         int, int, int, int, size_t, size_t, int);
void launch_mfma_probe_i8_test(const int8_t*, const int8_t*, int32_t*);
}

PYBIND11_MODULE(_core, m) {
    m.doc() = "crowdllama-amd MI355X (gfx950) HIP inference engine";

    m.def("nccl_unique_id", [] {
        ncclUniqueId id;
        if (ncclGetUniqueId(&id) != ncclSuccess)
            throw std::runtime_error("ncclGetUniqueId failed");
        return py::bytes(reinterpret_cast<const char*>(&id), sizeof(id));
    });

    m.def("device_count", [] {
        int n = 0;
        if (hipGetDeviceCount(&n) != hipSuccess) return 0;
        return n;
    });
    m.def("device_props", [](int dev) {
        hipDeviceProp_t p;
        HIP_CHECK(hipGetDeviceProperties(&p, dev));
        py::dict d;
        d["name"] = std::string(p.gcnArchName);
        d["total_mem_gb"] = (double)p.totalGlobalMem / (1024.0 * 1024 * 1024);
        d["multiprocessors"] = p.multiProcessorCount;
        d["warp_size"] = p.warpSize;
        return d;
    });

    struct PyTokenizer {
        Tokenizer* t;
        PyTokenizer(std::vector<std::string> tokens,
                    std::vector<std::string> merges, int bos, int eos)
            : t(tokenizer_new(std::move(tokens), merges, bos, eos)) {}
        ~PyTokenizer() { tokenizer_free(t); }
        PyTokenizer(const PyTokenizer&) = delete;
    };
    py::class_<PyTokenizer>(m, "Tokenizer")
        .def(py::init<std::vector<std::string>, std::vector<std::string>,
                      int, int>(),
             py::arg("tokens"), py::arg("merges") = std::vector<std::string>{},
             py::arg("bos_id") = 1, py::arg("eos_id") = 2)
        .def("encode",
             [](const PyTokenizer& pt, const std::string& s, bool add_bos) {
                 return tokenizer_encode(pt.t, s, add_bos);
             }, py::arg("text"), py::arg("add_bos") = true)
        .def("decode",
             [](const PyTokenizer& pt, std::vector<int32_t> ids) {
                 return py::bytes(tokenizer_decode(pt.t, ids));
             })
        .def_property_readonly("bos_id",
             [](const PyTokenizer& pt) { return tokenizer_bos(pt.t); })
        .def_property_readonly("eos_id",
             [](const PyTokenizer& pt) { return tokenizer_eos(pt.t); })
        .def("__len__",
             [](const PyTokenizer& pt) { return tokenizer_size(pt.t); });

    py::class_<EngineConfig>(m, "EngineConfig")
        .def(py::init<>())
        .def_readwrite("batch", &EngineConfig::batch)
        .def_readwrite("max_seq", &EngineConfig::max_seq)
        .def_readwrite("page_size", &EngineConfig::page_size)
        .def_readwrite("gen_cap", &EngineConfig::gen_cap)
        .def_readwrite("device", &EngineConfig::device)
        .def_readwrite("use_graph", &EngineConfig::use_graph)
        .def_readwrite("prefill_chunk", &EngineConfig::prefill_chunk)
        .def_readwrite("tp_rank", &EngineConfig::tp_rank)
        .def_readwrite("tp_size", &EngineConfig::tp_size)
        .def_readwrite("act_q8", &EngineConfig::act_q8)
        .def_readwrite("gemv_q8", &EngineConfig::gemv_q8)
        .def_property("nccl_id",
            [](EngineConfig& c) { return py::bytes(c.nccl_id); },
            [](EngineConfig& c, py::bytes b) { c.nccl_id = std::string(b); });

    py::class_<Engine>(m, "Engine")
        .def(py::init<const std::string&, const EngineConfig&>(),
             py::arg("gguf_path"), py::arg("config"))
        .def("reset", &Engine::reset)
        .def("reset_slot", &Engine::reset_slot)
        .def("set_slot_active", &Engine::set_slot_active)
        .def("prefill",
             [](Engine& e, py::array_t<int32_t, py::array::c_style> ids) {
                 if (ids.ndim() != 2)
                     throw std::runtime_error("ids must be [batch][len]");
                 const int len = (int)ids.shape(1);
                 std::vector<int32_t> v(ids.data(), ids.data() + ids.size());
                 py::gil_scoped_release rel;
                 e.prefill(v, len);
             })
        .def("decode", [](Engine& e, int n) {
                 py::gil_scoped_release rel;
                 e.decode(n);
             })
        .def("prefill_slot",
             [](Engine& e, int slot, std::vector<int32_t> ids) {
                 py::gil_scoped_release rel;
                 e.prefill_slot(slot, ids);
             })
        .def("gen_tokens", &Engine::gen_tokens)
        .def("logits",
             [](Engine& e, int slot) {
                 auto v = e.logits(slot);
                 return py::array_t<float>((py::ssize_t)v.size(), v.data());
             })
        .def("set_cur_token", &Engine::set_cur_token)
        .def("cur_token", &Engine::cur_token)
        .def("n_past", &Engine::n_past)
        .def("vram_bytes", &Engine::vram_bytes)
        .def("last_decode_ms", &Engine::last_decode_ms)
        .def_property_readonly("meta", [](Engine& e) {
            const auto& mm = e.meta();
            py::dict d;
            d["name"] = mm.name;
            d["vocab"] = mm.vocab;
            d["hidden"] = mm.hidden;
            d["layers"] = mm.layers;
            d["heads"] = mm.heads;
            d["kv_heads"] = mm.kv_heads;
            d["ffn"] = mm.ffn;
            d["head_dim"] = mm.head_dim;
            d["rope_theta"] = mm.rope_theta;
            d["rms_eps"] = mm.rms_eps;
            return d;
        });

    m.def("test_f16_decode", [](py::array_t<uint16_t> bits) {
        // host f16->f32 decoder (common.h) — CPU-testable
        py::array_t<float> out((py::ssize_t)bits.size());
        for (py::ssize_t i = 0; i < bits.size(); i++)
            out.mutable_data()[i] = f16_bits_to_f32_host(bits.data()[i]);
        return out;
    });
    m.def("test_rccl_graph_1rank", [](py::array_t<float> in) {
        std::vector<float> v(in.data(), in.data() + in.size());
        auto out = test_rccl_graph_1rank(v);
        return py::array_t<float>((py::ssize_t)out.size(), out.data());
    });
    m.def("test_gemm_i8", [](py::array_t<uint8_t> qs, py::array_t<uint8_t> hdr,
                             py::array_t<float, py::array::c_style> x,
                             int dtype, int N, int K, int force_splitk) {
        const int M = (int)x.shape(0);
        py::array_t<float> y({M, N});
        launch_gemm_i8_test(qs.data(), hdr.data(), x.data(),
                            y.mutable_data(), dtype, M, N, K, qs.nbytes(),
                            hdr.nbytes(), force_splitk);
        return y;
    }, py::arg("qs"), py::arg("hdr"), py::arg("x"), py::arg("dtype"),
       py::arg("N"), py::arg("K"), py::arg("force_splitk") = 0);
    m.def("test_mfma_probe_i8", [](py::array_t<int8_t> A,
                                   py::array_t<int8_t> B) {
        py::array_t<int32_t> C({16, 16});
        launch_mfma_probe_i8_test(A.data(), B.data(), C.mutable_data());
        return C;
    });
    m.def("test_gemm", [](py::array_t<uint8_t> qs, py::array_t<uint8_t> hdr,
                          py::array_t<float, py::array::c_style> x, int dtype,
                          int N, int K) {
        const int M = (int)x.shape(0);
        py::array_t<float> y({M, N});
        launch_gemm_test(qs.data(), hdr.data(), x.data(), y.mutable_data(),
                         dtype, M, N, K, qs.nbytes(), hdr.nbytes());
        return y;
    });
    m.def("bench_gemm", [](int dtype, int M, int N, int K, int iters) {
        return bench_gemm(dtype, M, N, K, iters);
    });
    m.def("test_slice_cols", [](int ggml_type, py::array_t<uint8_t> raw,
                                int rows, int k, int c0, int c1) {
        auto out = slice_cols_test(ggml_type, raw.data(), rows, k, c0, c1);
        return py::bytes(reinterpret_cast<const char*>(out.data()), out.size());
    });

    m.def("bench_membw", [](int nt, int mb, int wgs, int iters) {
        return bench_membw(nt, mb, wgs, iters);
    }, py::arg("nt") = 1, py::arg("mb") = 1024, py::arg("wgs") = 2048,
       py::arg("iters") = 10);
    m.def("bench_gemv_g", [](int dtype, int N, int K, int B, int iters) {
        return bench_gemv_g(dtype, N, K, B, iters);
    });
    m.def("bench_gemv_q8", [](int dtype, int N, int K, int B, int iters) {
        return bench_gemv_q8(dtype, N, K, B, iters);
    });
    m.def("test_gemv_q8", [](py::array_t<uint8_t> qs, py::array_t<uint8_t> hdr,
                             py::array_t<float, py::array::c_style> x,
                             int dtype, int N, int K) {
        const int B = (int)x.shape(0);
        py::array_t<float> y({B, N});
        launch_gemv_q8_test(qs.data(), hdr.data(), x.data(), y.mutable_data(),
                            dtype, N, K, B, qs.nbytes(), hdr.nbytes());
        return y;
    });
    m.def("bench_gemv", [](int dtype, int N, int K, int B, int pre, int iters) {
        // random weight bytes (content irrelevant for timing)
        const DT dt = static_cast<DT>(dtype);
        const size_t qs_bytes = (size_t)N * dqs_row_bytes(dt, K);
        const size_t hdr_bytes = (size_t)N * dhdr_row_bytes(dt, K);
        std::vector<uint8_t> qs(qs_bytes, 1), hdr(hdr_bytes ? hdr_bytes : 1, 1);
        return bench_gemv(qs.data(), hdr.data(), dtype, N, K, B, pre,
                          qs_bytes, hdr_bytes, iters);
    });

    m.def("test_mfma_probe", [](py::array_t<uint16_t> A, py::array_t<uint16_t> B) {
        py::array_t<float> C({16, 16});
        launch_mfma_probe_test(A.data(), B.data(), C.mutable_data());
        return C;
    });

    // ---- raw kernel entry points for numerics tests (tests/test_gpu_kernels.py)
    m.def("test_gemv", [](py::array_t<uint8_t> qs, py::array_t<uint8_t> hdr,
                          py::array_t<float> x, int dtype, int N, int K,
                          int pre, py::array_t<float> gw) {
        // x: [B][K] (or [B][2K] for PRE_SILU); returns y [B][N]
        const int B = (int)x.shape(0);
        const int xk = (int)x.shape(1);
        py::array_t<float> y({B, N});
        launch_gemv_test(qs.data(), hdr.data(), x.data(),
                         gw.size() ? gw.data() : nullptr, y.mutable_data(),
                         dtype, N, K, B, pre, qs.nbytes(), hdr.nbytes());
        (void)xk;
        return y;
    });
}
