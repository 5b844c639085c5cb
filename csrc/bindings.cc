// Python bindings for torch_cgx_amd._C.
//
// Surface parity with the reference pybind module
// (/root/reference/src/ProcessGroupCGX.cc:852-857): register_layer,
// set_quantization_bits, set_quantization_bucket_size (the reference's
// set_quantization_bucket_size mistakenly called SetQBits — fixed here),
// plus the ProcessGroupCGX class for the Python-side backend creator and
// direct kernel entry points used by the GPU tests.
#include <torch/extension.h>

#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <torch/csrc/utils/pybind.h>

#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>

#include <thread>

#include "backend.h"
#include "engine.h"
#include "transport.h"

namespace cgx {
namespace {

// Drive the REAL engine code at world size ws on ONE GPU: `buckets[r]` is
// rank r's flat bucket; ws Engine instances (each with its own streams,
// staging and events — exactly the production objects) exchange compressed
// chunks through the loopback transport (device-to-device copies with full
// stream/event fencing).  This is the hardware test seam the multi-rank
// SRA/Ring path runs under `pytest -m gpu` with a single MI355X: RCCL
// itself refuses two ranks on one device.
void run_loopback(const std::vector<at::Tensor>& tensors,
                  const std::function<hipStream_t(Engine&, Transport*,
                                                  hipStream_t, int)>& body) {
  const int ws = (int)tensors.size();
  TORCH_CHECK(ws >= 2, "loopback: need >= 2 per-rank tensors");
  for (const auto& t : tensors) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "loopback: CUDA contiguous tensors required");
    TORCH_CHECK(t.device() == tensors[0].device() &&
                    t.numel() == tensors[0].numel() &&
                    t.scalar_type() == tensors[0].scalar_type(),
                "loopback: tensors must match in device/numel/dtype");
  }
  const int dev = tensors[0].device().index();
  c10::hip::HIPGuardMasqueradingAsCUDA guard(dev);
  LoopbackHub hub(ws);
  std::vector<std::unique_ptr<Engine>> engines;
  std::vector<std::unique_ptr<LoopbackTransport>> trs;
  std::vector<hipStream_t> qs(ws, nullptr);
  for (int r = 0; r < ws; r++) {
    engines.push_back(std::make_unique<Engine>(r, ws));
    trs.push_back(std::make_unique<LoopbackTransport>(&hub, r));
    CGX_HIP_CHECK(hipStreamCreateWithFlags(&qs[r], hipStreamNonBlocking));
  }
  std::vector<std::exception_ptr> errs(ws);
  {
    py::gil_scoped_release nogil;
    std::vector<std::thread> th;
    for (int r = 0; r < ws; r++) {
      th.emplace_back([&, r] {
        try {
          c10::hip::HIPGuardMasqueradingAsCUDA g(dev);
          hipStream_t fin = body(*engines[r], trs[r].get(), qs[r], r);
          CGX_HIP_CHECK(hipStreamSynchronize(fin));
          CGX_HIP_CHECK(hipStreamSynchronize(qs[r]));
          CGX_HIP_CHECK(hipStreamSynchronize(engines[r]->comm_stream()));
          CGX_HIP_CHECK(hipStreamSynchronize(engines[r]->deq_stream()));
        } catch (...) {
          errs[r] = std::current_exception();
          hub.abort();  // unblock peers so join() cannot hang
        }
      });
    }
    for (auto& t : th) t.join();
  }
  (void)hipDeviceSynchronize();  // loopback events/copies fully retired
  for (auto s : qs)
    if (s) (void)hipStreamDestroy(s);
  for (int r = 0; r < ws; r++)
    if (errs[r]) std::rethrow_exception(errs[r]);
}

void py_loopback_allreduce(std::vector<at::Tensor> buckets) {
  const int64_t numel = buckets.empty() ? 0 : buckets[0].numel();
  // resolve the registry ONCE and force it into every engine, exactly like
  // the hierarchical path (one cursor step per bucket across all ranks)
  Registry::BucketInfo info;
  const bool matched = Registry::get().next(numel, nullptr, &info);
  run_loopback(buckets, [&](Engine& e, Transport* tr, hipStream_t qs, int r) {
    return e.allreduce(buckets[r], tr, qs, &info, matched);
  });
}

// Multiple allreduce steps on PERSISTENT engines (one set of engines for
// all steps, like a training run): steps[s][r] is rank r's bucket at step
// s, reduced in place.  This is what lets error-feedback residuals carry
// across steps on hardware.
void py_loopback_allreduce_multi(
    std::vector<std::vector<at::Tensor>> steps) {
  TORCH_CHECK(!steps.empty(), "loopback_allreduce_multi: no steps");
  const int ws = (int)steps[0].size();
  const int64_t numel = steps[0][0].numel();
  for (auto& st : steps) {
    TORCH_CHECK((int)st.size() == ws && st[0].numel() == numel,
                "loopback_allreduce_multi: ragged steps");
  }
  Registry::BucketInfo info;
  const bool matched = Registry::get().next(numel, nullptr, &info);
  run_loopback(steps[0],
               [&](Engine& e, Transport* tr, hipStream_t qs, int r) {
                 hipStream_t fin = qs;
                 for (size_t s = 0; s < steps.size(); s++) {
                   fin = e.allreduce(steps[s][r], tr, qs, &info, matched);
                 }
                 return fin;
               });
}

void py_loopback_broadcast(std::vector<at::Tensor> tensors, int64_t root) {
  run_loopback(tensors, [&](Engine& e, Transport* tr, hipStream_t qs, int r) {
    return e.broadcast(tensors[r], (int)root, tr, qs);
  });
}

// Hierarchical flow on one GPU (backend.cc ProcessGroupCGX::allreduce
// `hier` branch): ws = n_nodes * local_size ranks; each "node" gets an
// intra hub, the node leaders share a cross hub.  Per rank: intra-node
// compressed allreduce, cross-node reduction on the leader, intra
// broadcast of the result — the registry resolved ONCE for all engines.
void py_loopback_hierarchical(std::vector<at::Tensor> buckets,
                              int64_t local_size) {
  const int ws = (int)buckets.size();
  TORCH_CHECK(local_size >= 1 && ws % local_size == 0 && ws >= 2,
              "loopback_hierarchical: world must be n_nodes*local_size");
  const int nodes = ws / (int)local_size;
  TORCH_CHECK(nodes >= 2, "loopback_hierarchical: need >= 2 nodes");
  for (const auto& t : buckets) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                    t.numel() == buckets[0].numel() &&
                    t.device() == buckets[0].device(),
                "loopback_hierarchical: matching CUDA tensors required");
  }
  const int dev = buckets[0].device().index();
  c10::hip::HIPGuardMasqueradingAsCUDA guard(dev);
  Registry::BucketInfo info;
  const bool matched =
      Registry::get().next(buckets[0].numel(), nullptr, &info);
  std::vector<std::unique_ptr<LoopbackHub>> intra_hubs;
  for (int nd = 0; nd < nodes; nd++)
    intra_hubs.push_back(std::make_unique<LoopbackHub>((int)local_size));
  LoopbackHub cross_hub(nodes);
  std::vector<std::unique_ptr<Engine>> intra_eng;    // per rank
  std::vector<std::unique_ptr<LoopbackTransport>> intra_tr;
  std::vector<std::unique_ptr<Engine>> cross_eng;    // per node (leader)
  std::vector<std::unique_ptr<LoopbackTransport>> cross_tr;
  std::vector<hipStream_t> qs(ws, nullptr);
  for (int r = 0; r < ws; r++) {
    const int nd = r / (int)local_size;
    const int lr = r % (int)local_size;
    intra_eng.push_back(std::make_unique<Engine>(lr, (int)local_size));
    intra_tr.push_back(
        std::make_unique<LoopbackTransport>(intra_hubs[nd].get(), lr));
    CGX_HIP_CHECK(hipStreamCreateWithFlags(&qs[r], hipStreamNonBlocking));
  }
  for (int nd = 0; nd < nodes; nd++) {
    cross_eng.push_back(
        std::make_unique<Engine>(nd, nodes, /*is_cross=*/true));
    cross_tr.push_back(std::make_unique<LoopbackTransport>(&cross_hub, nd));
  }
  std::vector<std::exception_ptr> errs(ws);
  {
    py::gil_scoped_release nogil;
    std::vector<std::thread> th;
    for (int r = 0; r < ws; r++) {
      th.emplace_back([&, r] {
        try {
          c10::hip::HIPGuardMasqueradingAsCUDA g(dev);
          const int nd = r / (int)local_size;
          const int lr = r % (int)local_size;
          hipStream_t fin = intra_eng[r]->allreduce(
              buckets[r], intra_tr[r].get(), qs[r], &info, matched);
          if (lr == 0) {
            fin = cross_eng[nd]->allreduce(buckets[r], cross_tr[nd].get(),
                                           fin, &info, matched);
          }
          fin = intra_eng[r]->broadcast(buckets[r], /*root=*/0,
                                        intra_tr[r].get(), fin);
          CGX_HIP_CHECK(hipStreamSynchronize(fin));
          CGX_HIP_CHECK(hipStreamSynchronize(qs[r]));
        } catch (...) {
          errs[r] = std::current_exception();
          for (auto& h : intra_hubs) h->abort();
          cross_hub.abort();
        }
      });
    }
    for (auto& t : th) t.join();
  }
  (void)hipDeviceSynchronize();
  for (auto s : qs)
    if (s) (void)hipStreamDestroy(s);
  for (int r = 0; r < ws; r++)
    if (errs[r]) std::rethrow_exception(errs[r]);
}

DType dtype_arg(const at::Tensor& t) { return dtype_of(t); }

// Compress a 1-D CUDA tensor; returns the uint8 compressed buffer
// (zero-filled alignment padding so byte comparisons are well-defined).
at::Tensor py_quantize(at::Tensor x, int64_t bits, int64_t bucket_size,
                       bool stochastic, int64_t seed, bool skip_incomplete,
                       c10::optional<at::Tensor> feedback) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "quantize: CUDA contiguous");
  TORCH_CHECK(bits >= 1 && bits <= 8, "quantize: bits in [1,8]");
  TORCH_CHECK(bucket_size >= 1, "quantize: bucket_size >= 1");
  void* fbp = nullptr;
  if (feedback.has_value()) {
    TORCH_CHECK(feedback->is_cuda() && feedback->is_contiguous() &&
                    feedback->numel() == x.numel() &&
                    feedback->scalar_type() == x.scalar_type(),
                "quantize: feedback must match x");
    TORCH_CHECK(bucket_size % 8 == 0,
                "quantize: error feedback requires bucket_size % 8 == 0");
    fbp = feedback->data_ptr();
  }
  const DType dt = dtype_arg(x);
  const int64_t n = x.numel();
  const int64_t bytes =
      buffer_size(n, dt, (int)bits, (int)bucket_size, skip_incomplete);
  auto out = at::zeros({std::max<int64_t>(bytes, 1)},
                       at::TensorOptions().dtype(at::kByte).device(x.device()));
  if (n == 0) return out;

  struct Blob {
    QuantDesc d;
    int64_t cum[2];
  } hb;
  auto stream =
      c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(x.device().index());
  auto upload = [&](const Blob& blob) {
    return at::from_blob(const_cast<Blob*>(&blob), {(int64_t)sizeof(Blob)},
                         at::TensorOptions().dtype(at::kByte))
        .to(x.device());
  };
  const int32_t base_flags = skip_incomplete ? kFlagSkipIncomplete : 0;
  const bool aligned =
      (reinterpret_cast<uintptr_t>(x.data_ptr()) & 15) == 0;
  // mirror the engine's launch split (run_quantize): lean fast kernel for
  // the full buckets, generic kernel for the partial tail
  if (bucket_size % 8 == 0 && !fbp && aligned && n >= bucket_size &&
      bucket_size <= 2048) {  // register-stash limit (4 groups/lane)
    hb.d = QuantDesc{x.data_ptr(), out.data_ptr<uint8_t>(), nullptr, n,
                     (int32_t)bucket_size, base_flags};
    hb.cum[0] = 0;
    hb.cum[1] = n / bucket_size;
    auto dev = upload(hb);
    const char* devp = static_cast<const char*>(dev.data_ptr());
    const int64_t ng = bucket_size >> 3;
    if (ng < 64 && (ng & (ng - 1)) == 0) {
      launch_quantize_sub(
          reinterpret_cast<const QuantDesc*>(devp),
          reinterpret_cast<const int64_t*>(devp + offsetof(Blob, cum)), 1,
          hb.cum[1], dt, (int)bits, (uint64_t)seed, stochastic,
          stream.stream(), skip_incomplete && (n % bucket_size) != 0);
    } else {
      launch_quantize_fast(
          reinterpret_cast<const QuantDesc*>(devp),
          reinterpret_cast<const int64_t*>(devp + offsetof(Blob, cum)), 1,
          hb.cum[1], dt, (int)bits, (uint64_t)seed, stochastic,
          stream.stream(), skip_incomplete && (n % bucket_size) != 0,
          (int)((bucket_size / 8 + 63) / 64));
    }
    if (!skip_incomplete && (n % bucket_size) != 0) {
      Blob tb;
      tb.d = QuantDesc{x.data_ptr(), out.data_ptr<uint8_t>(), nullptr, n,
                       (int32_t)bucket_size, base_flags | kFlagTailOnly};
      tb.cum[0] = 0;
      tb.cum[1] = 1;
      auto tdev = upload(tb);
      const char* tdevp = static_cast<const char*>(tdev.data_ptr());
      launch_quantize_batch(
          reinterpret_cast<const QuantDesc*>(tdevp),
          reinterpret_cast<const int64_t*>(tdevp + offsetof(Blob, cum)), 1, 1,
          dt, (int)bits, (uint64_t)seed, stochastic, stream.stream(),
          /*buckets_mult8=*/true, false);
    }
    return out;
  }
  hb.d = QuantDesc{x.data_ptr(), out.data_ptr<uint8_t>(), fbp, n,
                   (int32_t)bucket_size, base_flags};
  hb.cum[0] = 0;
  hb.cum[1] = skip_incomplete ? n / bucket_size
                              : (n + bucket_size - 1) / bucket_size;
  auto dev = upload(hb);
  const char* devp = static_cast<const char*>(dev.data_ptr());
  launch_quantize_batch(
      reinterpret_cast<const QuantDesc*>(devp),
      reinterpret_cast<const int64_t*>(devp + offsetof(Blob, cum)), 1,
      hb.cum[1], dt, (int)bits, (uint64_t)seed, stochastic, stream.stream(),
      bucket_size % 8 == 0,
      skip_incomplete && (n % bucket_size) != 0);
  return out;
}

// Decompress `comp` into `out` (1-D CUDA tensor of n elements); add=True
// accumulates in T precision.
void py_dequantize(at::Tensor comp, at::Tensor out, int64_t bits,
                   int64_t bucket_size, bool add, bool skip_incomplete) {
  TORCH_CHECK(comp.is_cuda() && out.is_cuda() && out.is_contiguous());
  const DType dt = dtype_arg(out);
  const int64_t n = out.numel();
  if (n == 0) return;
  const int64_t nq =
      skip_incomplete ? n / bucket_size * bucket_size : n;
  struct Blob {
    DequantDesc d;
    int64_t cum[2];
  } hb;
  auto stream =
      c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(out.device().index());
  auto upload = [&](const Blob& blob) {
    return at::from_blob(const_cast<Blob*>(&blob), {(int64_t)sizeof(Blob)},
                         at::TensorOptions().dtype(at::kByte))
        .to(out.device());
  };
  const int32_t base_flags = skip_incomplete ? kFlagSkipIncomplete : 0;
  hb.d = DequantDesc{comp.data_ptr<uint8_t>(), out.data_ptr(), n, 0,
                     (int32_t)bucket_size, 1, add ? 1 : 0, base_flags};
  hb.cum[0] = 0;
  hb.cum[1] = (nq + 7) / 8;
  // mirror the engine's launch split (run_dequant)
  if (bucket_size % 8 == 0 && nq < (int64_t(1) << 31)) {
    auto dev = upload(hb);
    const char* devp = static_cast<const char*>(dev.data_ptr());
    launch_dequantize_fast(reinterpret_cast<const DequantDesc*>(devp), 1,
                           hb.cum[1], dt, (int)bits, stream.stream(),
                           skip_incomplete && (n % bucket_size) != 0);
    const bool ragged =
        elem_size(dt) == 4 ? (nq & 3) != 0 : (nq & 7) != 0;
    if (ragged) {
      Blob tb = hb;
      tb.d.flags = base_flags | kFlagTailOnly;
      tb.cum[1] = 1;
      auto tdev = upload(tb);
      const char* tdevp = static_cast<const char*>(tdev.data_ptr());
      launch_dequantize_batch(
          reinterpret_cast<const DequantDesc*>(tdevp),
          reinterpret_cast<const int64_t*>(tdevp + offsetof(Blob, cum)), 1, 1,
          dt, (int)bits, stream.stream(), false);
    }
    return;
  }
  auto dev = upload(hb);
  const char* devp = static_cast<const char*>(dev.data_ptr());
  launch_dequantize_batch(
      reinterpret_cast<const DequantDesc*>(devp),
      reinterpret_cast<const int64_t*>(devp + offsetof(Blob, cum)), 1,
      hb.cum[1], dt, (int)bits, stream.stream(),
      skip_incomplete && (n % bucket_size) != 0);
}

// Multi-source decode-and-sum: comp is [nsrc, stride] uint8; decodes each
// row's stream and accumulates in T precision (the engine's round-1 path).
void py_dequantize_multi(at::Tensor comp, at::Tensor out, int64_t bits,
                         int64_t bucket_size, bool add) {
  TORCH_CHECK(comp.is_cuda() && comp.dim() == 2 && comp.is_contiguous());
  TORCH_CHECK(out.is_cuda() && out.is_contiguous());
  const DType dt = dtype_arg(out);
  const int64_t n = out.numel();
  if (n == 0) return;
  struct Blob {
    DequantDesc d;
    int64_t cum[2];
  } hb;
  hb.d = DequantDesc{comp.data_ptr<uint8_t>(), out.data_ptr(), n,
                     comp.stride(0), (int32_t)bucket_size,
                     (int32_t)comp.size(0), add ? 1 : 0, 0};
  hb.cum[0] = 0;
  hb.cum[1] = (n + 7) / 8;
  auto dev = at::from_blob(&hb, {(int64_t)sizeof(Blob)},
                           at::TensorOptions().dtype(at::kByte))
                 .to(out.device());
  auto stream =
      c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(out.device().index());
  const char* devp = static_cast<const char*>(dev.data_ptr());
  if (bucket_size % 8 == 0 && n < (int64_t(1) << 31)) {
    launch_dequantize_fast(reinterpret_cast<const DequantDesc*>(devp), 1,
                           hb.cum[1], dt, (int)bits, stream.stream());
    const bool ragged = elem_size(dt) == 4 ? (n & 3) != 0 : (n & 7) != 0;
    if (ragged) {
      Blob tb = hb;
      tb.d.flags = kFlagTailOnly;
      tb.cum[1] = 1;
      auto tdev = at::from_blob(&tb, {(int64_t)sizeof(Blob)},
                                at::TensorOptions().dtype(at::kByte))
                      .to(out.device());
      const char* tdevp = static_cast<const char*>(tdev.data_ptr());
      launch_dequantize_batch(
          reinterpret_cast<const DequantDesc*>(tdevp),
          reinterpret_cast<const int64_t*>(tdevp + offsetof(Blob, cum)), 1, 1,
          dt, (int)bits, stream.stream());
    }
    return;
  }
  launch_dequantize_batch(
      reinterpret_cast<const DequantDesc*>(devp),
      reinterpret_cast<const int64_t*>(devp + offsetof(Blob, cum)), 1,
      hb.cum[1], dt, (int)bits, stream.stream());
}

int64_t py_buffer_size(int64_t n, at::ScalarType st, int64_t bits,
                       int64_t bucket_size, bool skip_incomplete) {
  DType dt = st == at::kFloat ? DType::F32
             : st == at::kHalf ? DType::F16
                               : DType::BF16;
  return buffer_size(n, dt, (int)bits, (int)bucket_size, skip_incomplete);
}

std::pair<std::vector<int64_t>, std::vector<int64_t>> py_partition(
    int64_t num_elements, int64_t world_size,
    std::vector<int64_t> layer_numels, int64_t esize) {
  std::vector<int64_t> offs, szs;
  Engine::partition(num_elements, (int)world_size, layer_numels, (int)esize,
                    &offs, &szs);
  return {offs, szs};
}

void py_register_layer(int64_t bucket_idx, int64_t layer_idx, int64_t numel,
                       int64_t bits, int64_t bucket_size) {
  Registry::get().register_layer((int)bucket_idx, (int)layer_idx, numel,
                                 (int)bits, (int)bucket_size);
}

void py_set_bits(int64_t bucket_idx, int64_t layer_idx, int64_t bits) {
  Registry::get().set_bits((int)bucket_idx, (int)layer_idx, (int)bits);
}

void py_set_bucket(int64_t bucket_idx, int64_t layer_idx,
                   int64_t bucket_size) {
  Registry::get().set_bucket_size((int)bucket_idx, (int)layer_idx,
                                  (int)bucket_size);
}

}  // namespace
}  // namespace cgx

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  py::class_<cgx::ProcessGroupCGX,
             c10::intrusive_ptr<cgx::ProcessGroupCGX>, c10d::Backend>(
      m, "ProcessGroupCGX")
      .def(py::init<c10::intrusive_ptr<c10d::Store>, int, int,
                    c10::intrusive_ptr<c10d::Backend>>(),
           py::arg("store"), py::arg("rank"), py::arg("size"),
           py::arg("cpu_delegate") = nullptr);

  m.def("register_layer", &cgx::py_register_layer, py::arg("bucket_idx"),
        py::arg("layer_idx"), py::arg("layer_numel"), py::arg("bits"),
        py::arg("bucket_size"));
  m.def("set_quantization_bits", &cgx::py_set_bits);
  m.def("set_quantization_bucket_size", &cgx::py_set_bucket);
  m.def("clear_registry", [] { cgx::Registry::get().clear(); });
  m.def("registry_match",
        [](int64_t numel, int64_t key) {
          // test seam for Registry::next: key is an opaque bucket-storage
          // identity (the engine passes the flat tensor's data_ptr)
          cgx::Registry::BucketInfo info;
          const bool ok = cgx::Registry::get().next(
              numel, reinterpret_cast<const void*>(key), &info);
          return py::make_tuple(ok, ok ? info.idx : -1);
        },
        py::arg("numel"), py::arg("key") = 0);
  m.def("registry_snapshot", [] {
    // [(bucket_idx, [numels], [(bits, bucket_size)])] in registration order
    py::list out;
    auto& reg = cgx::Registry::get();
    for (const auto& b : reg.snapshot()) {
      py::list numels, cfgs;
      for (auto n : b.numels) numels.append(n);
      for (const auto& c : b.cfgs)
        cfgs.append(py::make_tuple(c.bits, c.bucket_size));
      out.append(py::make_tuple(b.idx, numels, cfgs));
    }
    return out;
  });

  m.def("parse_engine_config", [] {
    // test seam: the env-derived engine configuration, as each Engine
    // re-reads it per bucket (inner_ring/cross_ring select the reduction
    // algorithm for the intra-node vs cross-node engine independently)
    auto c = cgx::EngineConfig::from_env();
    py::dict d;
    d["fusion_bytes"] = c.fusion_bytes;
    d["min_elems"] = c.min_elems;
    d["default_bits"] = c.default_bits;
    d["default_bucket"] = c.default_bucket;
    d["stochastic"] = c.stochastic;
    d["inner_ring"] = c.ring;
    d["cross_ring"] = c.cross_ring;
    d["debug_a2a"] = c.debug_a2a;
    d["fake_ratio"] = c.fake_ratio;
    d["skip_incomplete"] = c.skip_incomplete;
    d["dummy"] = c.dummy;
    d["intra_compress"] = c.intra_compress;
    d["error_feedback"] = c.error_feedback;
    return d;
  });
  m.def("loopback_allreduce", &cgx::py_loopback_allreduce, py::arg("buckets"),
        "Run the production Engine::allreduce at world_size=len(buckets) on "
        "one GPU via the loopback transport; buckets[r] is rank r's flat "
        "bucket tensor, reduced in place.");
  m.def("loopback_broadcast", &cgx::py_loopback_broadcast, py::arg("tensors"),
        py::arg("root") = 0,
        "Run the production Engine::broadcast at world_size=len(tensors) on "
        "one GPU via the loopback transport.");
  m.def("loopback_allreduce_multi", &cgx::py_loopback_allreduce_multi,
        py::arg("steps"),
        "Multiple allreduce steps on persistent engines (steps[s][r] = rank "
        "r's bucket at step s, reduced in place); error-feedback residuals "
        "carry across steps.");
  m.def("loopback_hierarchical", &cgx::py_loopback_hierarchical,
        py::arg("buckets"), py::arg("local_size"),
        "Run the production hierarchical flow (intra allreduce -> leader "
        "cross reduction -> intra broadcast) for n_nodes*local_size ranks "
        "on one GPU via loopback transports.");
  m.def("quantize", &cgx::py_quantize, py::arg("x"), py::arg("bits"),
        py::arg("bucket_size"), py::arg("stochastic") = false,
        py::arg("seed") = 0, py::arg("skip_incomplete") = false,
        py::arg("feedback") = py::none());
  m.def("dequantize", &cgx::py_dequantize, py::arg("comp"), py::arg("out"),
        py::arg("bits"), py::arg("bucket_size"), py::arg("add") = false,
        py::arg("skip_incomplete") = false);
  m.def("dequantize_multi", &cgx::py_dequantize_multi, py::arg("comp"),
        py::arg("out"), py::arg("bits"), py::arg("bucket_size"),
        py::arg("add") = false);
  m.def("compute_topology",
        [](const c10::intrusive_ptr<c10d::Store>& store, int rank, int size,
           const std::string& hostname) {
          auto t = cgx::compute_topology(store, rank, size, hostname);
          return std::make_tuple(t.node_id, t.local_rank, t.local_size,
                                 t.n_nodes, t.uniform);
        },
        "Exchange hostnames through the store -> (node_id, local_rank, "
        "local_size, n_nodes, uniform)");
  m.def("buffer_size", &cgx::py_buffer_size, py::arg("n"), py::arg("dtype"),
        py::arg("bits"), py::arg("bucket_size"),
        py::arg("skip_incomplete") = false);
  m.def("partition", &cgx::py_partition);
}
