// pybind11 bindings for the adapcc_amd native engine.
//
// Deliberately torch-free at the C++ level: tensors arrive as
// (data_ptr, numel, dtype) and the HIP stream as an integer
// (torch.cuda.current_stream().cuda_stream), so the extension builds with
// plain hipcc and the Python wrapper (adapcc_amd/runtime/engine.py) owns all
// torch-level safety checks.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "common.h"
#include "engine.h"
#include "plan.h"

namespace py = pybind11;
using adapcc::Engine;

// Expose the unit-plan computation (plan.cpp) for CPU-side unit tests: the
// Python test harness simulates the full multi-rank protocol over these
// plans with numpy buffers, covering relay control and deadlock-freedom
// without a GPU.
static py::dict plan_to_dict(const adapcc::PlanData& p) {
  py::list cs, rs, bs;
  for (const auto& u : p.cunits) {
    py::dict d;
    d["tree"] = u.tree; d["chunk"] = u.chunk;
    d["offset"] = u.offset_elems; d["count"] = u.count_elems;
    d["flag_space"] = (int)u.flag_space;
    py::list nl;
    for (int i = 0; i < u.nnotify; ++i) nl.append(u.notify_rank[i]);
    d["notify_to"] = nl;
    cs.append(d);
  }
  for (const auto& u : p.runits) {
    py::dict d;
    d["tree"] = u.tree; d["chunk"] = u.chunk;
    d["offset"] = u.offset_elems; d["count"] = u.count_elems;
    py::list srcs;
    for (int s = 0; s < u.nsrc; ++s)
      srcs.append(py::make_tuple(u.src_rank[s], (int)u.src_kind[s]));
    d["srcs"] = srcs;
    d["include_self"] = (bool)u.include_self;
    d["notify"] = (bool)u.notify_parent; d["consumer"] = u.parent_rank;
    d["is_root"] = (bool)u.is_root;
    py::list kids;
    for (int c = 0; c < u.nchildren; ++c) kids.append(u.child_rank[c]);
    d["publish_to"] = kids;
    rs.append(d);
  }
  for (const auto& u : p.bunits) {
    py::dict d;
    d["tree"] = u.tree; d["chunk"] = u.chunk;
    d["src_offset"] = u.src_offset_elems; d["dst_offset"] = u.dst_offset_elems;
    d["count"] = u.count_elems;
    d["parent"] = u.parent_rank; d["parent_kind"] = (int)u.parent_kind;
    d["forward"] = (bool)u.forward;
    py::list kids;
    for (int c = 0; c < u.nchildren; ++c) kids.append(u.child_rank[c]);
    d["publish_to"] = kids;
    bs.append(d);
  }
  py::dict out;
  out["copy"] = cs;
  out["reduce"] = rs;
  out["bcast"] = bs;
  out["chunk_elems"] = p.chunk_elems;
  return out;
}

namespace adapcc {
void launch_local_reduce(Dtype dt, void* dst, const void* const* srcs_dev,
                         int nsrc, long count, RedOp op, float scale,
                         hipStream_t stream);
bool ln_supported_api(long cols, int dtype);
void ln_forward(int dtype, const void* x, const void* w, const void* b,
                void* y, float* mean, float* rstd, long rows, long cols,
                float eps, hipStream_t stream);
void ln_backward(int dtype, const void* dy, const void* x, const void* w,
                 const float* mean, const float* rstd, void* dx,
                 float* ws_gamma, float* ws_beta, long rows, long cols,
                 int nblocks, hipStream_t stream);
void ce_forward(int dtype, const void* logits, const void* targets,
                float* loss, float* lse, long rows, long cols,
                long ignore_index, hipStream_t stream);
void ce_backward(int dtype, const void* logits, const void* targets,
                 const float* lse, const float* gscale, void* dlogits,
                 long rows, long cols, long ignore_index, hipStream_t stream);
void fa_forward(const void* q, const void* k, const void* v, void* o,
                float* lse, int B, int H, int S, const long* strides,
                float scale, hipStream_t stream);
void fa_backward(const void* q, const void* k, const void* v, const void* do_,
                 const float* lse, const float* delta, void* dq, void* dk,
                 void* dv, int B, int H, int S, const long* strides,
                 float scale, hipStream_t stream);
void mfma_probe(const void* a, const void* b, float* c, hipStream_t stream);
void gelu_forward(int dtype, const void* x, void* y, long n,
                  hipStream_t stream);
void gelu_backward(int dtype, const void* x, const void* dy, void* dx, long n,
                   hipStream_t stream);
void tr_probe(float* out, int kb, int db, hipStream_t stream);
void pack_probe(float* out, hipStream_t stream);
}

PYBIND11_MODULE(_core, m) {
  m.doc() = "adapcc_amd native engine (MI355X / gfx950)";

  // dst = op over srcs (local GPU buffers), scaled. Synchronous; used by
  // the single-GPU numerics tests and fused combine ops.
  m.def("local_reduce",
        [](uintptr_t dst, const std::vector<uintptr_t>& srcs, long count,
           int dtype, int op, float scale, uintptr_t stream) {
          if (srcs.empty()) throw std::runtime_error("no sources");
          void** d_srcs = nullptr;
          hipError_t e = hipMalloc(&d_srcs, srcs.size() * sizeof(void*));
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
          e = hipMemcpy(d_srcs, srcs.data(), srcs.size() * sizeof(void*),
                        hipMemcpyHostToDevice);
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
          adapcc::launch_local_reduce(
              (adapcc::Dtype)dtype, reinterpret_cast<void*>(dst),
              (const void* const*)d_srcs, (int)srcs.size(), count,
              (adapcc::RedOp)op, scale, reinterpret_cast<hipStream_t>(stream));
          e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
          e = hipStreamSynchronize(reinterpret_cast<hipStream_t>(stream));
          (void)hipFree(d_srcs);
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        },
        py::arg("dst"), py::arg("srcs"), py::arg("count"), py::arg("dtype"),
        py::arg("op"), py::arg("scale") = 1.0f, py::arg("stream") = 0);

  m.def("ln_supported", &adapcc::ln_supported_api, py::arg("cols"),
        py::arg("dtype"));
  m.def("ln_fwd",
        [](int dtype, uintptr_t x, uintptr_t w, uintptr_t b, uintptr_t y,
           uintptr_t mean, uintptr_t rstd, long rows, long cols, float eps,
           uintptr_t stream) {
          adapcc::ln_forward(dtype, (const void*)x, (const void*)w,
                             (const void*)b, (void*)y, (float*)mean,
                             (float*)rstd, rows, cols, eps,
                             reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });
  m.def("ln_bwd",
        [](int dtype, uintptr_t dy, uintptr_t x, uintptr_t w, uintptr_t mean,
           uintptr_t rstd, uintptr_t dx, uintptr_t wsg, uintptr_t wsb,
           long rows, long cols, int nblocks, uintptr_t stream) {
          adapcc::ln_backward(dtype, (const void*)dy, (const void*)x,
                              (const void*)w, (const float*)mean,
                              (const float*)rstd, (void*)dx, (float*)wsg,
                              (float*)wsb, rows, cols, nblocks,
                              reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });

  m.def("ce_fwd",
        [](int dtype, uintptr_t logits, uintptr_t targets, uintptr_t loss,
           uintptr_t lse, long rows, long cols, long ignore_index,
           uintptr_t stream) {
          adapcc::ce_forward(dtype, (const void*)logits,
                             (const void*)targets, (float*)loss, (float*)lse,
                             rows, cols, ignore_index,
                             reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });
  m.def("ce_bwd",
        [](int dtype, uintptr_t logits, uintptr_t targets, uintptr_t lse,
           uintptr_t gscale, uintptr_t dlogits, long rows, long cols,
           long ignore_index, uintptr_t stream) {
          adapcc::ce_backward(dtype, (const void*)logits,
                              (const void*)targets, (const float*)lse,
                              (const float*)gscale, (void*)dlogits, rows,
                              cols, ignore_index,
                              reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });

  m.def("fa_fwd",
        [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t o, uintptr_t lse,
           int B, int H, int S, const std::vector<long>& strides, float scale,
           uintptr_t stream) {
          if (strides.size() != 12)
            throw std::runtime_error("fa_fwd: need 12 strides");
          adapcc::fa_forward((const void*)q, (const void*)k, (const void*)v,
                             (void*)o, (float*)lse, B, H, S, strides.data(),
                             scale, reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });
  m.def("fa_bwd",
        [](uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t dout,
           uintptr_t lse, uintptr_t delta, uintptr_t dq, uintptr_t dk,
           uintptr_t dv, int B, int H, int S, const std::vector<long>& strides,
           float scale, uintptr_t stream) {
          if (strides.size() != 21)
            throw std::runtime_error("fa_bwd: need 21 strides");
          adapcc::fa_backward((const void*)q, (const void*)k, (const void*)v,
                              (const void*)dout, (const float*)lse,
                              (const float*)delta, (void*)dq, (void*)dk,
                              (void*)dv, B, H, S, strides.data(), scale,
                              reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });
  m.def("mfma_probe",
        [](uintptr_t a, uintptr_t b, uintptr_t c, uintptr_t stream) {
          adapcc::mfma_probe((const void*)a, (const void*)b, (float*)c,
                             reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });

  m.def("gelu_fwd",
        [](int dtype, uintptr_t x, uintptr_t y, long n, uintptr_t stream) {
          adapcc::gelu_forward(dtype, (const void*)x, (void*)y, n,
                               reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });
  m.def("gelu_bwd",
        [](int dtype, uintptr_t x, uintptr_t dy, uintptr_t dx, long n,
           uintptr_t stream) {
          adapcc::gelu_backward(dtype, (const void*)x, (const void*)dy,
                                (void*)dx, n,
                                reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });

  m.def("tr_probe",
        [](uintptr_t out, int kb, int db, uintptr_t stream) {
          adapcc::tr_probe((float*)out, kb, db,
                           reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });
  m.def("pack_probe",
        [](uintptr_t out, uintptr_t stream) {
          adapcc::pack_probe((float*)out,
                             reinterpret_cast<hipStream_t>(stream));
          hipError_t e = hipGetLastError();
          if (e != hipSuccess) throw std::runtime_error(hipGetErrorString(e));
        });

  m.def("compute_plan",
        [](const std::vector<std::vector<int>>& parents, int rank,
           long total_elems, int esize, long chunk_bytes,
           const std::vector<int>& active,
           const std::vector<double>& slice_weights) {
          auto shape = adapcc::TreeShape::derive(parents);
          uint64_t mask = 0;
          for (int r : active) mask |= (1ull << r);
          if (active.empty()) mask = (1ull << shape.world) - 1;
          return plan_to_dict(adapcc::build_plan(shape, rank, total_elems,
                                                 esize, chunk_bytes, mask,
                                                 slice_weights));
        },
        py::arg("parents"), py::arg("rank"), py::arg("total_elems"),
        py::arg("esize"), py::arg("chunk_bytes"),
        py::arg("active") = std::vector<int>{},
        py::arg("slice_weights") = std::vector<double>{});

  m.def("compute_primitive_plan",
        [](const std::string& prim, int world, int rank, long elems, int esize,
           long chunk_bytes, int root,
           const std::vector<std::vector<int>>& parents,
           const std::vector<int>& active,
           const std::vector<double>& slice_weights) {
          uint64_t mask = 0;
          for (int r : active) mask |= (1ull << r);
          if (active.empty()) mask = (1ull << world) - 1;
          adapcc::PlanData pd;
          if (prim == "reduce") {
            auto shape = parents.empty() ? adapcc::star_shape(world)
                                         : adapcc::TreeShape::derive(parents);
            pd = adapcc::build_reduce_plan(shape, rank, root, elems, esize,
                                           chunk_bytes, mask, slice_weights);
          } else if (prim == "broadcast") {
            pd = adapcc::build_broadcast_plan(world, rank, root, elems, esize,
                                              chunk_bytes);
          } else if (prim == "allgather") {
            pd = adapcc::build_allgather_plan(world, rank, elems, esize,
                                              chunk_bytes);
          } else if (prim == "reducescatter") {
            pd = adapcc::build_reducescatter_plan(world, rank, elems, esize,
                                                  chunk_bytes, mask);
          } else if (prim == "alltoall") {
            pd = adapcc::build_alltoall_plan(world, rank, elems, esize,
                                             chunk_bytes);
          } else {
            throw std::runtime_error("unknown primitive " + prim);
          }
          return plan_to_dict(pd);
        },
        py::arg("prim"), py::arg("world"), py::arg("rank"), py::arg("elems"),
        py::arg("esize"), py::arg("chunk_bytes"), py::arg("root") = 0,
        py::arg("parents") = std::vector<std::vector<int>>{},
        py::arg("active") = std::vector<int>{},
        py::arg("slice_weights") = std::vector<double>{});

  py::class_<Engine>(m, "Engine")
      .def(py::init<int, int, int, size_t, double>(), py::arg("rank"),
           py::arg("world"), py::arg("device"), py::arg("cap_bytes"),
           py::arg("timeout_ms"))
      .def("ipc_handle",
           [](Engine& e) { return py::bytes(e.ipc_handle()); })
      .def("connect",
           [](Engine& e, const std::vector<py::bytes>& handles,
              const std::vector<int>& peer_devices) {
             std::vector<std::string> hs;
             hs.reserve(handles.size());
             for (const auto& h : handles) hs.push_back(std::string(h));
             e.connect(hs, peer_devices);
           },
           py::arg("handles"), py::arg("peer_devices") = std::vector<int>{})
      .def("set_strategy", &Engine::set_strategy, py::arg("parents"),
           py::arg("chunk_bytes"),
           py::arg("slice_weights") = std::vector<double>{})
      .def("connect_local", &Engine::connect_local, py::arg("peer_addrs"))
      .def("region_addr", &Engine::region_addr)
      .def("allreduce",
           [](Engine& e, uintptr_t data_ptr, long numel, int dtype, int op,
              const std::vector<int>& active, bool average,
              uintptr_t stream) {
             e.allreduce(reinterpret_cast<void*>(data_ptr), numel, dtype, op,
                         active, average, reinterpret_cast<void*>(stream));
           },
           py::arg("data_ptr"), py::arg("numel"), py::arg("dtype"),
           py::arg("op"), py::arg("active"), py::arg("average"),
           py::arg("stream"))
      .def("reduce",
           [](Engine& e, uintptr_t data_ptr, long numel, int dtype, int op,
              int root, const std::vector<int>& active, uintptr_t stream) {
             e.reduce(reinterpret_cast<void*>(data_ptr), numel, dtype, op,
                      root, active, reinterpret_cast<void*>(stream));
           },
           py::arg("data_ptr"), py::arg("numel"), py::arg("dtype"),
           py::arg("op"), py::arg("root"), py::arg("active"),
           py::arg("stream"))
      .def("broadcast",
           [](Engine& e, uintptr_t data_ptr, long numel, int dtype, int root,
              uintptr_t stream) {
             e.broadcast(reinterpret_cast<void*>(data_ptr), numel, dtype, root,
                         reinterpret_cast<void*>(stream));
           },
           py::arg("data_ptr"), py::arg("numel"), py::arg("dtype"),
           py::arg("root"), py::arg("stream"))
      .def("all_gather",
           [](Engine& e, uintptr_t in_ptr, uintptr_t out_ptr, long in_elems,
              int dtype, uintptr_t stream) {
             e.all_gather(reinterpret_cast<const void*>(in_ptr),
                          reinterpret_cast<void*>(out_ptr), in_elems, dtype,
                          reinterpret_cast<void*>(stream));
           },
           py::arg("in_ptr"), py::arg("out_ptr"), py::arg("in_elems"),
           py::arg("dtype"), py::arg("stream"))
      .def("all_to_all",
           [](Engine& e, uintptr_t in_ptr, uintptr_t out_ptr, long per_rank,
              int dtype, uintptr_t stream) {
             e.all_to_all(reinterpret_cast<const void*>(in_ptr),
                          reinterpret_cast<void*>(out_ptr), per_rank, dtype,
                          reinterpret_cast<void*>(stream));
           },
           py::arg("in_ptr"), py::arg("out_ptr"), py::arg("per_rank_elems"),
           py::arg("dtype"), py::arg("stream"))
      .def("reduce_scatter",
           [](Engine& e, uintptr_t in_ptr, uintptr_t out_ptr, long out_elems,
              int dtype, int op, const std::vector<int>& active, bool average,
              uintptr_t stream) {
             e.reduce_scatter(reinterpret_cast<const void*>(in_ptr),
                              reinterpret_cast<void*>(out_ptr), out_elems,
                              dtype, op, active, average,
                              reinterpret_cast<void*>(stream));
           },
           py::arg("in_ptr"), py::arg("out_ptr"), py::arg("out_elems"),
           py::arg("dtype"), py::arg("op"), py::arg("active"),
           py::arg("average"), py::arg("stream"))
      .def("synchronize", &Engine::synchronize,
           py::call_guard<py::gil_scoped_release>())
      .def("query_error", &Engine::query_error)
      .def("dump_inbox", [](Engine& e) { return py::bytes(e.dump_inbox()); })
      .def_property_readonly("rank", &Engine::rank)
      .def_property_readonly("world", &Engine::world)
      .def_property_readonly("capacity", &Engine::capacity)
      .def_property_readonly("num_trees", &Engine::num_trees);

  m.attr("DTYPE_F32") = (int)adapcc::Dtype::F32;
  m.attr("DTYPE_F16") = (int)adapcc::Dtype::F16;
  m.attr("DTYPE_BF16") = (int)adapcc::Dtype::BF16;
  m.attr("OP_SUM") = (int)adapcc::RedOp::Sum;
  m.attr("OP_AVG") = (int)adapcc::RedOp::Avg;
  m.attr("OP_MAX") = (int)adapcc::RedOp::Max;
  m.attr("OP_MIN") = (int)adapcc::RedOp::Min;
  m.attr("MAX_RANKS") = adapcc::kMaxRanks;
  m.attr("MAX_TREES") = adapcc::kMaxTrees;
}
