// Python bindings for accl_amd — the PyACCL-equivalent surface.
// Buffers interoperate zero-copy with torch via DLPack (__dlpack__), so the
// bench path touches no host memory.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>
#include <cstring>
#include <unistd.h>
#include "../core/accl.hpp"
#include "../emu/emudevice.hpp"
#include <hip/hip_runtime.h>
#include "../gpu/gpudevice.hpp"

namespace py = pybind11;
using namespace accl;

// ------------------------------- minimal DLPack (no external header) ------
namespace dlpack {
struct DLDevice { int32_t device_type; int32_t device_id; };
struct DLDataType { uint8_t code; uint8_t bits; uint16_t lanes; };
struct DLTensor {
  void* data; DLDevice device; int32_t ndim; DLDataType dtype;
  int64_t* shape; int64_t* strides; uint64_t byte_offset;
};
struct DLManagedTensor {
  DLTensor dl_tensor; void* manager_ctx;
  void (*deleter)(DLManagedTensor*);
};
constexpr int kDLCPU = 1, kDLROCM = 10;
}  // namespace dlpack

static dlpack::DLDataType dl_dtype(DataType dt) {
  switch (dt) {
    case DataType::float32: return {2, 32, 1};
    case DataType::float64: return {2, 64, 1};
    case DataType::float16: return {2, 16, 1};
    case DataType::bfloat16: return {4, 16, 1};
    case DataType::int32: return {0, 32, 1};
    case DataType::int64: return {0, 64, 1};
    case DataType::int8: return {0, 8, 1};
    default: return {2, 32, 1};
  }
}

struct DLHolder {
  int64_t shape[1];
  dlpack::DLManagedTensor mt;
};

static py::capsule make_dlpack(BaseBuffer& b, bool gpu, int device_id) {
  auto* h = new DLHolder();
  h->shape[0] = int64_t(b.count());
  h->mt.dl_tensor.data = b.device_ptr();
  h->mt.dl_tensor.device = {gpu ? dlpack::kDLROCM : dlpack::kDLCPU,
                            gpu ? device_id : 0};
  h->mt.dl_tensor.ndim = 1;
  h->mt.dl_tensor.dtype = dl_dtype(b.dtype());
  h->mt.dl_tensor.shape = h->shape;
  h->mt.dl_tensor.strides = nullptr;
  h->mt.dl_tensor.byte_offset = 0;
  h->mt.manager_ctx = h;
  h->mt.deleter = [](dlpack::DLManagedTensor* t) {
    delete (DLHolder*)t->manager_ctx;
  };
  return py::capsule(&h->mt, "dltensor", [](PyObject* cap) {
    if (PyCapsule_IsValid(cap, "dltensor")) {
      auto* mt = (dlpack::DLManagedTensor*)PyCapsule_GetPointer(cap, "dltensor");
      if (mt && mt->deleter) mt->deleter(mt);
    }
  });
}

// ------------------------------------------------------------ module ----
PYBIND11_MODULE(_core, m) {
  m.doc() = "accl_amd core: MI355X-native ACCL-class collective engine";

  py::enum_<DataType>(m, "DataType")
      .value("none", DataType::none)
      .value("float16", DataType::float16)
      .value("float32", DataType::float32)
      .value("float64", DataType::float64)
      .value("int32", DataType::int32)
      .value("int64", DataType::int64)
      .value("bfloat16", DataType::bfloat16)
      .value("int8", DataType::int8);

  py::enum_<Op>(m, "Op")
      .value("copy", Op::copy)
      .value("combine", Op::combine)
      .value("send", Op::send)
      .value("recv", Op::recv)
      .value("bcast", Op::bcast)
      .value("scatter", Op::scatter)
      .value("gather", Op::gather)
      .value("reduce", Op::reduce)
      .value("allgather", Op::allgather)
      .value("allreduce", Op::allreduce)
      .value("reduce_scatter", Op::reduce_scatter)
      .value("alltoall", Op::alltoall)
      .value("barrier", Op::barrier)
      .value("stream_put", Op::stream_put);

  py::enum_<ReduceFunction>(m, "ReduceFunction")
      .value("SUM", ReduceFunction::SUM)
      .value("MAX", ReduceFunction::MAX);

  m.attr("TAG_ANY") = py::int_(u64(TAG_ANY));
  m.attr("GLOBAL_COMM") = py::int_(GLOBAL_COMM);

  py::class_<BaseBuffer>(m, "Buffer")
      .def_property_readonly("count", &BaseBuffer::count)
      .def_property_readonly("bytes", &BaseBuffer::bytes)
      .def_property_readonly("dtype", &BaseBuffer::dtype)
      .def_property_readonly("arena_offset", &BaseBuffer::arena_offset)
      .def("sync_to_device", &BaseBuffer::sync_to_device)
      .def("sync_from_device", &BaseBuffer::sync_from_device)
      .def("slice",
           [](BaseBuffer& b, u64 start, u64 end) { return b.slice(start, end); })
      .def("write",
           [](BaseBuffer& b, py::buffer data, bool to_device) {
             py::buffer_info info = data.request();
             u64 n = u64(info.size) * u64(info.itemsize);
             if (n > b.bytes()) throw std::runtime_error("write too large");
             if (b.host_ptr()) std::memcpy(b.host_ptr(), info.ptr, n);
             if (to_device) {
               if (b.host_ptr()) b.sync_to_device();
             }
           },
           py::arg("data"), py::arg("to_device") = true)
      .def("read",
           [](BaseBuffer& b, py::buffer out, bool from_device) {
             py::buffer_info info = out.request(true);
             u64 n = u64(info.size) * u64(info.itemsize);
             if (n > b.bytes()) throw std::runtime_error("read too large");
             if (from_device) b.sync_from_device();
             std::memcpy(info.ptr, b.host_ptr(), n);
           },
           py::arg("out"), py::arg("from_device") = true);

  py::class_<Request>(m, "Request")
      .def("wait", &Request::wait, py::arg("timeout_ms") = 120000,
           py::call_guard<py::gil_scoped_release>())
      .def("test", &Request::test)
      .def("retcode", &Request::retcode)
      .def("duration_us", &Request::duration_us,
           py::call_guard<py::gil_scoped_release>());

  py::class_<ACCL>(m, "ACCL")
      .def_property_readonly("rank", &ACCL::rank)
      .def_property_readonly("nranks", &ACCL::nranks)
      .def("local_blob",
           [](ACCL& a) {
             auto b = a.local_blob();
             return py::bytes(b.data(), b.size());
           })
      .def("connect",
           [](ACCL& a, const std::vector<py::bytes>& blobs) {
             std::vector<std::vector<char>> v;
             for (auto& b : blobs) {
               std::string s = b;
               v.emplace_back(s.begin(), s.end());
             }
             a.connect(v);
           },
           py::call_guard<py::gil_scoped_release>())
      .def("deinit", &ACCL::deinit, py::call_guard<py::gil_scoped_release>())
      .def("debug_wave_tiles",
           [](ACCL& a) {
             auto* g = dynamic_cast<GpuDevice*>(a.backend());
             return g ? g->debug_wave_tiles() : std::vector<u32>{};
           })
      .def("debug_timeline",
           [](ACCL& a) {
             auto* g = dynamic_cast<GpuDevice*>(a.backend());
             return g ? g->debug_timeline() : std::vector<u64>{};
           })
      .def("create_buffer",
           [](ACCL& a, u64 count, DataType dt, bool device_only) {
             return device_only ? a.create_buffer_device(count, dt)
                                : a.create_buffer(count, dt);
           },
           py::arg("count"), py::arg("dtype"), py::arg("device_only") = false,
           py::keep_alive<0, 1>())
      .def("buffer_dlpack",
           [](ACCL& a, BaseBuffer& b, int device_id) {
             return make_dlpack(b, a.backend()->is_gpu(), device_id);
           },
           py::arg("buffer"), py::arg("device_id") = 0)
      .def("put", &ACCL::put, py::arg("src"), py::arg("count"),
           py::arg("dst_rank"), py::arg("peer_arena_offset"),
           py::arg("from_device") = false, py::arg("run_async") = false,
           py::call_guard<py::gil_scoped_release>(),
           py::return_value_policy::reference)
      .def("create_communicator", &ACCL::create_communicator)
      .def("split_communicator", &ACCL::split_communicator)
      .def("copy", &ACCL::copy, py::arg("src"), py::arg("dst"), py::arg("count"),
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("combine", &ACCL::combine, py::arg("count"), py::arg("function"),
           py::arg("op0"), py::arg("op1"), py::arg("result"),
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("send", &ACCL::send, py::arg("src"), py::arg("count"), py::arg("dst"),
           py::arg("tag") = u32(TAG_ANY), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("recv", &ACCL::recv, py::arg("dst"), py::arg("count"), py::arg("src"),
           py::arg("tag") = u32(TAG_ANY), py::arg("comm") = GLOBAL_COMM,
           py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("bcast", &ACCL::bcast, py::arg("buf"), py::arg("count"),
           py::arg("root"), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("scatter", &ACCL::scatter, py::arg("src"), py::arg("dst"),
           py::arg("count"), py::arg("root"), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("gather", &ACCL::gather, py::arg("src"), py::arg("dst"),
           py::arg("count"), py::arg("root"), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("allgather", &ACCL::allgather, py::arg("src"), py::arg("dst"),
           py::arg("count"), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("reduce", &ACCL::reduce, py::arg("src"), py::arg("dst"),
           py::arg("count"), py::arg("root"), py::arg("function"),
           py::arg("comm") = GLOBAL_COMM, py::arg("from_device") = false,
           py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("allreduce", &ACCL::allreduce, py::arg("src"), py::arg("dst"),
           py::arg("count"), py::arg("function"), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("reduce_scatter", &ACCL::reduce_scatter, py::arg("src"),
           py::arg("dst"), py::arg("count"), py::arg("function"),
           py::arg("comm") = GLOBAL_COMM, py::arg("from_device") = false,
           py::arg("to_device") = false,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("alltoall", &ACCL::alltoall, py::arg("src"), py::arg("dst"),
           py::arg("count"), py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false, py::arg("to_device") = false,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("stream_put", &ACCL::stream_put, py::arg("src"), py::arg("count"),
           py::arg("dst"), py::arg("tag") = 0, py::arg("comm") = GLOBAL_COMM,
           py::arg("from_device") = false,
           py::arg("compress") = DataType::none, py::arg("run_async") = false,
           py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("pop_stream",
           [](ACCL& a, u32 src, py::buffer out, u64 timeout_ms) {
             py::buffer_info info = out.request(true);
             u64 maxb = u64(info.size) * u64(info.itemsize);
             u32 tag = 0;
             u64 n;
             {
               py::gil_scoped_release rel;
               n = a.pop_stream(src, info.ptr, maxb, &tag, timeout_ms);
             }
             return py::make_tuple(n, tag);
           },
           py::arg("src"), py::arg("out"), py::arg("timeout_ms") = 10000)
      .def("stream_ready", &ACCL::stream_ready, py::arg("src"))
      .def("push_stream",
           [](ACCL& a, u32 dst, py::buffer data, u32 tag, u64 timeout_ms) {
             py::buffer_info info = data.request();
             return a.push_stream(dst, info.ptr,
                                  u64(info.size) * u64(info.itemsize), tag,
                                  timeout_ms);
           },
           py::arg("dst"), py::arg("data"), py::arg("tag") = 0,
           py::arg("timeout_ms") = 10000,
           py::call_guard<py::gil_scoped_release>())
      .def("copy_from_stream", &ACCL::copy_from_stream, py::arg("lane"),
           py::arg("dst"), py::arg("count"), py::arg("to_device") = false,
           py::arg("run_async") = false, py::return_value_policy::reference,
           py::keep_alive<0, 1>(), py::call_guard<py::gil_scoped_release>())
      .def("send_from_stream", &ACCL::send_from_stream, py::arg("lane"),
           py::arg("count"), py::arg("dst"), py::arg("tag") = u32(TAG_ANY),
           py::arg("comm") = GLOBAL_COMM,
           py::arg("compress_dtype") = DataType::none,
           py::arg("run_async") = false, py::return_value_policy::reference,
           py::keep_alive<0, 1>(), py::call_guard<py::gil_scoped_release>())
      .def("alive", &ACCL::alive)
      .def("soft_reset", &ACCL::soft_reset,
           py::call_guard<py::gil_scoped_release>())
      .def("info",
           [](ACCL& a) {
             // reference: parse_hwid capability decode (accl.cpp:1050-1064)
             const ProtoConfig& c = a.backend()->cfg();
             py::dict d;
             d["backend"] = a.backend()->is_gpu() ? "gpu" : "emu";
             d["rank"] = c.rank;
             d["nranks"] = c.nranks;
             d["n_eager_slots"] = c.n_slots;
             d["eager_slot_bytes"] = c.slot_bytes;
             d["n_stream_slots"] = c.n_stream;
             d["stream_slot_bytes"] = c.stream_bytes;
             d["max_eager_bytes"] = c.max_eager;
             d["timeout_us"] = c.timeout_us;
             return d;
           })
      .def("set_timeout_ms", &ACCL::set_timeout_ms,
           py::call_guard<py::gil_scoped_release>())
      .def("set_tuning", &ACCL::set_tuning, py::arg("knob"),
           py::arg("value"), py::call_guard<py::gil_scoped_release>())
      .def("set_max_rendezvous_size", &ACCL::set_max_rendezvous_size,
           py::call_guard<py::gil_scoped_release>())
      .def("set_max_eager_size", &ACCL::set_max_eager_size,
           py::call_guard<py::gil_scoped_release>())
      .def("dump_communicator", &ACCL::dump_communicator,
           py::arg("comm") = GLOBAL_COMM)
      .def("dump_eager_rx_buffers", &ACCL::dump_eager_rx_buffers,
           py::arg("verbose") = false)
      .def("dump_streams", &ACCL::dump_streams)
      .def("dump_engine_status", &ACCL::dump_engine_status)
      .def("dump_rendezvous", &ACCL::dump_rendezvous)
      .def("barrier", &ACCL::barrier, py::arg("comm") = GLOBAL_COMM,
           py::arg("run_async") = false, py::return_value_policy::reference, py::keep_alive<0, 1>(),
           py::call_guard<py::gil_scoped_release>())
      .def("nop", &ACCL::nop, py::arg("run_async") = false,
           py::return_value_policy::reference)
      .def("free_request", &ACCL::free_request);

  auto mk_cfg = [](u32 nranks, u32 rank, py::dict opts) {
    ProtoConfig c = default_proto_config(nranks, rank);
    if (opts.contains("n_slots")) c.n_slots = opts["n_slots"].cast<u32>();
    if (opts.contains("slot_bytes")) c.slot_bytes = opts["slot_bytes"].cast<u32>();
    if (opts.contains("max_eager")) c.max_eager = opts["max_eager"].cast<u64>();
    if (opts.contains("timeout_us")) c.timeout_us = opts["timeout_us"].cast<u64>();
    if (opts.contains("n_stream")) c.n_stream = opts["n_stream"].cast<u32>();
    if (opts.contains("stream_bytes")) c.stream_bytes = opts["stream_bytes"].cast<u32>();
    if (opts.contains("n_rndzv")) c.n_rndzv = opts["n_rndzv"].cast<u32>();
    return c;
  };

  m.def("create_emu",
        [mk_cfg](u32 nranks, u32 rank, const std::string& job, u64 heap_bytes,
                 py::dict opts) {
          ProtoConfig c = mk_cfg(nranks, rank, opts);
          auto be = std::unique_ptr<Backend>(
              new EmuDevice(nranks, rank, job, &c, heap_bytes));
          return new ACCL(std::move(be));
        },
        py::arg("nranks"), py::arg("rank"), py::arg("job"),
        py::arg("heap_bytes") = u64(256u << 20), py::arg("opts") = py::dict());

  m.def("create_gpu",
        [mk_cfg](u32 nranks, u32 rank, int device, u64 heap_bytes, int wgs,
                 py::dict opts) {
          ProtoConfig c = mk_cfg(nranks, rank, opts);
          auto be = std::unique_ptr<Backend>(
              new GpuDevice(nranks, rank, device, &c, heap_bytes, wgs));
          return new ACCL(std::move(be));
        },
        py::arg("nranks"), py::arg("rank"), py::arg("device") = 0,
        py::arg("heap_bytes") = u64(8ull << 30), py::arg("wgs") = 0,
        py::arg("opts") = py::dict());

  m.def("error_to_string", &error_to_string);

  // Raw-pointer collective call: operands are DEVICE pointers (e.g. torch
  // tensors' data_ptr), not arena buffers — zero staging copies. Eager
  // schedules only touch local src/dst (peers go through arena slots), so
  // raw pointers are legal; direct paths require arena residency and the
  // engine falls back to eager when the arena flags are absent.
  m.def("call_raw",
        [](ACCL& a, u32 scenario, u64 count, u32 root, u32 tag, u32 comm,
           u32 function, u64 addr0, u64 addr1, u64 addr2, u32 dtype,
           u32 wire_dtype, bool run_async) {
          CallDesc d{};
          d.scenario = scenario;
          d.count_lo = u32(count);
          d.count_hi = u32(count >> 32);
          d.root_src_dst = root;
          d.tag = tag;
          d.comm_id = comm;
          d.function = function;
          d.addr0 = addr0;
          d.addr1 = addr1;
          d.addr2 = addr2;
          d.flags = 0;  // raw pointers: no arena flags
          d.arith = dtype | (wire_dtype << 8);
          Backend* be = a.backend();
          u64 seq = be->submit(d);
          auto* r = new Request(be, seq);
          if (!run_async) {
            u32 e = r->wait();
            if (e) {
              delete r;
              std::string msg = "accl call_raw failed: " + error_to_string(e);
              if (e & E_TIMEOUT) msg += "\n" + be->timeout_dump_str();
              throw accl_error(msg, e);
            }
          }
          return r;
        },
        py::arg("a"), py::arg("scenario"), py::arg("count"),
        py::arg("root") = 0, py::arg("tag") = 0, py::arg("comm") = 0,
        py::arg("function") = 0, py::arg("addr0") = 0, py::arg("addr1") = 0,
        py::arg("addr2") = 0, py::arg("dtype") = u32(DataType::float32),
        py::arg("wire_dtype") = u32(DataType::float32),
        py::arg("run_async") = false,
        py::call_guard<py::gil_scoped_release>(),
        py::return_value_policy::take_ownership);

  // deployment introspection (reference: xclbin_scan enumerating kernels /
  // memory banks, driver/utils/xclbin_scan)
  m.def("device_info", []() {
    py::list out;
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return out;
    for (int i = 0; i < n; ++i) {
      hipDeviceProp_t p{};
      if (hipGetDeviceProperties(&p, i) != hipSuccess) continue;
      py::dict d;
      d["index"] = i;
      d["name"] = std::string(p.name);
      d["gcn_arch"] = std::string(p.gcnArchName);
      d["total_mem_gb"] = double(p.totalGlobalMem) / (1u << 30);
      d["multiprocessors"] = p.multiProcessorCount;
      d["xgmi_capable"] = bool(p.isLargeBar);
      // peer matrix (the xclbin_scan deployment-introspection analogue,
      // reference driver/utils/xclbin_scan/): reachability + link perf
      // rank per peer device
      py::list peers;
      for (int j = 0; j < n; ++j) {
        if (j == i) continue;
        int can = 0;
        (void)hipDeviceCanAccessPeer(&can, i, j);
        py::dict pd;
        pd["device"] = j;
        pd["p2p"] = bool(can);
        int perf = 0;
        if (hipDeviceGetP2PAttribute(&perf, hipDevP2PAttrPerformanceRank, i,
                                     j) == hipSuccess)
          pd["perf_rank"] = perf;
        peers.append(pd);
      }
      d["peers"] = peers;
      out.append(d);
    }
    return out;
  });

  // engine capability word (the HWID/parse_hwid analogue, reference
  // driver/xrt/src/accl.cpp:1050-1064): what this build of the engine
  // supports, for deployment introspection
  m.def("engine_capabilities", []() {
    py::dict d;
    d["version"] = 2;
    d["max_ranks"] = MAX_RANKS;
    d["max_comms"] = MAX_COMMS;
    d["max_inflight_parked"] = MAX_INFLIGHT;
    d["move_ring"] = MOVE_RING;
    d["ring_cap"] = RING_CAP;
    d["devcall_ring"] = DEVCALL_RING;
    py::list ops;
    for (const char* s :
         {"copy", "combine", "send", "recv", "bcast", "scatter", "gather",
          "reduce", "allgather", "allreduce", "reduce_scatter", "alltoall",
          "barrier", "stream_put"})
      ops.append(std::string(s));
    d["ops"] = ops;
    py::list dts;
    for (const char* s :
         {"float16", "float32", "float64", "int32", "int64", "bfloat16",
          "int8"})
      dts.append(std::string(s));
    d["dtypes"] = dts;
    d["protocols"] = py::make_tuple("eager", "rendezvous_direct");
    d["features"] = py::make_tuple(
        "multi_call_interleaving", "ooo_rendezvous_matching",
        "windowed_nary_fan_in", "compression_f16_bf16_wire",
        "device_initiated_calls", "stream_rings", "torch_backend",
        "nonblocking_probe_ops", "parked_drain_progress",
        "flow_controlled_rndzv_rings", "pooled_progress_words",
        "one_shot_small_fan_in", "wildcard_user_tags");
    return d;
  });

  // host-side injector into the device-call ring: exercises the
  // client_arbiter path on the emulator (and doubles as a raw call API)
  // TEST-ONLY single-producer injection: emulates device_call() from the
  // host via read/modify/write of the ring head, which is NOT atomic
  // against concurrent device-side device_call() fetch_adds. Use only in
  // harnesses where no kernel produces device calls at the same time.
  m.def("inject_device_call",
        [](ACCL& a, u32 scenario, u64 count, u32 root, u32 tag, u64 addr0,
           u64 addr2, u32 flags, u32 function) {
          Backend* be = a.backend();
          const ProtoConfig& c = be->cfg();
          ArenaLayout L = arena_layout(c);
          DevCallRing ring{};
          be->read_arena(L.devcall_off, &ring, sizeof(ring));
          u64 idx = ring.head;
          ring.head = idx + 1;
          be->write_arena(L.devcall_off, &ring, sizeof(u64));
          if (idx >= DEVCALL_RING) {
            // slot-reuse guard (mirrors device_call): wait until the
            // previous occupant's ret was published by the engine
            u64 ret_off = L.devcall_off + sizeof(DevCallRing) +
                          u64(DEVCALL_RING) * sizeof(DevCallSlot) +
                          (idx % DEVCALL_RING) * sizeof(DevCallRet);
            u64 t0 = wallclock_host_ns(), prev_seq = 0;
            do {
              be->read_arena(ret_off, &prev_seq, sizeof(prev_seq));
              if (prev_seq >= idx + 1 - DEVCALL_RING) break;
              if (wallclock_host_ns() - t0 > 30ull * 1000000000)
                throw accl_error("inject_device_call: ring slot never freed");
              usleep(50);
            } while (true);
          }
          DevCallSlot slot{};
          slot.d.scenario = scenario;
          slot.d.count_lo = u32(count);
          slot.d.count_hi = u32(count >> 32);
          slot.d.root_src_dst = root;
          slot.d.tag = tag;
          slot.d.addr0 = addr0;
          slot.d.addr2 = addr2;
          slot.d.flags = flags;
          slot.d.function = function;
          slot.d.arith = u32(DataType::float32) | (u32(DataType::float32) << 8);
          u64 off = L.devcall_off + sizeof(DevCallRing) +
                    (idx % DEVCALL_RING) * sizeof(DevCallSlot);
          be->write_arena(off + offsetof(DevCallSlot, d), &slot.d,
                          sizeof(CallDesc));
          u64 seq = idx + 1;
          be->write_arena(off + offsetof(DevCallSlot, seq), &seq, sizeof(seq));
          return idx;
        })
  ;
  m.def("wait_device_call",
        [](ACCL& a, u64 token, u64 timeout_ms) {
          Backend* be = a.backend();
          ArenaLayout L = arena_layout(be->cfg());
          u64 off = L.devcall_off + sizeof(DevCallRing) +
                    u64(DEVCALL_RING) * sizeof(DevCallSlot) +
                    (token % DEVCALL_RING) * sizeof(DevCallRet);
          DevCallRet r{};
          u64 t0 = wallclock_host_ns();
          for (;;) {
            be->read_arena(off, &r, sizeof(r));
            if (r.seq >= token + 1) return u32(r.errcode);
            if (wallclock_host_ns() - t0 > timeout_ms * 1000000ull)
              throw accl_error("wait_device_call: timeout");
            cpu_pause();
          }
        },
        py::arg("a"), py::arg("token"), py::arg("timeout_ms") = 10000);

  m.def("demo_vadd_devicecall",
        [](ACCL& a, BaseBuffer& src, BaseBuffer& scratch, u64 count, u32 dst,
           u32 tag, float addv) {
          auto* g = dynamic_cast<GpuDevice*>(a.backend());
          if (!g) throw accl_error("demo_vadd_devicecall: gpu backend only");
          hipStream_t st{};
          if (hipStreamCreateWithFlags(&st, hipStreamNonBlocking) != hipSuccess)
            throw accl_error("demo: stream create failed");
          launch_vadd_devicecall(
              g->arena_local() + src.arena_offset(),
              g->arena_local() + scratch.arena_offset(),
              scratch.arena_offset(), count, tag, g->arena_local(), dst, addv,
              st);
          hipError_t e = hipGetLastError();
          u64 t0 = wallclock_host_ns();
          while (e == hipSuccess) {
            hipError_t q = hipStreamQuery(st);
            if (q != hipErrorNotReady) { e = q; break; }
            if (wallclock_host_ns() - t0 > 30ull * 1000000000) {
              e = hipErrorUnknown;
              break;
            }
            usleep(100);
          }
          (void)hipStreamDestroy(st);
          if (e != hipSuccess)
            throw accl_error(std::string("demo_vadd_devicecall: ") +
                             hipGetErrorString(e));
        },
        py::arg("a"), py::arg("src"), py::arg("scratch"), py::arg("count"),
        py::arg("dst"), py::arg("tag") = 0, py::arg("addv") = 1.0f,
        py::call_guard<py::gil_scoped_release>());

  // demo plugin: device-initiated stream_put (reference vadd_put analogue)
  m.def("demo_vadd_put",
        [](ACCL& a, BaseBuffer& src, u64 count, u32 dst, u32 tag, float addv) {
          auto* g = dynamic_cast<GpuDevice*>(a.backend());
          if (!g) throw accl_error("demo_vadd_put: gpu backend only");
          u32 seg = g->cfg().stream_bytes < (32u << 10) ? g->cfg().stream_bytes
                                                        : (32u << 10);
          // fresh stream: the engine's streams hold never-ending persistent
          // kernels, so anything queued behind them would never run
          hipStream_t st{};
          if (hipStreamCreateWithFlags(&st, hipStreamNonBlocking) != hipSuccess)
            throw accl_error("demo_vadd_put: stream create failed");
          launch_vadd_put(g->arena_local() + src.arena_offset(), count, tag,
                          g->arena_local(), g->peer_base(dst), g->cfg().rank,
                          dst, seg, addv, st);
          hipError_t e = hipGetLastError();
          u64 t0 = wallclock_host_ns();
          while (e == hipSuccess) {  // bounded sync (30 s)
            hipError_t q = hipStreamQuery(st);
            if (q != hipErrorNotReady) { e = q; break; }
            if (wallclock_host_ns() - t0 > 30ull * 1000000000) {
              e = hipErrorUnknown;
              break;
            }
            usleep(100);
          }
          (void)hipStreamDestroy(st);
          if (e != hipSuccess)
            throw accl_error(std::string("demo_vadd_put: ") +
                             hipGetErrorString(e));
        },
        py::arg("a"), py::arg("src"), py::arg("count"), py::arg("dst"),
        py::arg("tag") = 0, py::arg("addv") = 1.0f,
        py::call_guard<py::gil_scoped_release>());
}
