// pybind11 bindings: the Python client talks to the C++ engine through this
// module (parity role: scanner/engine/python.cpp). Job specs travel as
// msgpack bytes (packed with the msgpack wheel on the Python side, decoded
// by csrc/msgpack.h here).
#include <dlfcn.h>

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../kernels/dnn.h"
#include "dag/graph.h"
#include "engine/executor.h"
#include "hip_util.h"
#include "memory.h"
#include "msgpack.h"
#include "ops/source.h"
#include "video/h264.h"
#include "video/ingest.h"
#include "video/mp4.h"
#include "ops/kernel.h"
#include "ops/python_kernel.h"
#include "video/svc.h"

namespace py = pybind11;
using namespace sca;

namespace {

mp::Value mp_from_pybytes(const py::bytes& b) {
  std::string s = b;
  return mp::decode(reinterpret_cast<const u8*>(s.data()), s.size());
}

JobGraph graph_from_bytes(const py::bytes& b) {
  return JobGraph::from_msgpack(mp_from_pybytes(b));
}

std::vector<JobBinding> jobs_from_bytes(const py::bytes& b) {
  std::vector<JobBinding> jobs;
  mp::Value v = mp_from_pybytes(b);
  for (auto& jv : v.as_array()) {
    jobs.push_back(JobBinding::from_msgpack(jv));
  }
  return jobs;
}

PerfParams perf_from_dict(const py::dict& d) {
  PerfParams pp;
  if (d.contains("io_packet_size"))
    pp.io_packet_size = d["io_packet_size"].cast<i64>();
  if (d.contains("work_packet_size"))
    pp.work_packet_size = d["work_packet_size"].cast<i64>();
  if (d.contains("pipeline_instances"))
    pp.pipeline_instances = d["pipeline_instances"].cast<i32>();
  if (d.contains("load_workers"))
    pp.load_workers = d["load_workers"].cast<i32>();
  if (d.contains("cpu_pool_size"))
    pp.cpu_pool_size = d["cpu_pool_size"].cast<size_t>();
  if (d.contains("gpu_pool_size"))
    pp.gpu_pool_size = d["gpu_pool_size"].cast<size_t>();
  if (d.contains("sparsity_threshold"))
    pp.sparsity_threshold = d["sparsity_threshold"].cast<i32>();
  if (d.contains("profiler_level"))
    pp.profiler_level = d["profiler_level"].cast<i32>();
  if (d.contains("span_cache_size"))
    pp.span_cache_size = d["span_cache_size"].cast<size_t>();
  return pp;
}

py::list profilers_to_py(const std::vector<std::unique_ptr<Profiler>>& profs) {
  py::list out;
  for (auto& p : profs) {
    py::dict d;
    py::list iv;
    for (auto& i : p->intervals()) {
      iv.append(py::make_tuple(i.label, i.start_ns, i.end_ns));
    }
    d["intervals"] = iv;
    py::dict cnt;
    for (auto& kv : p->counters()) cnt[py::str(kv.first)] = kv.second;
    d["counters"] = cnt;
    out.append(d);
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "scanner_amd C++/HIP engine";

  register_stdlib_ops();
  register_gpu_ops();
  register_resnet50_op();
  register_optflow_gpu();
  register_pose_op();
  register_color_gpu();
  register_image_encoder_op();
  register_detector_op();
  register_files_source_sink();

  // Load a user op plugin .so built with tools/build_op.py (parity:
  // Client.load_op / REGISTER_OP static registrars in user libraries,
  // scannerpy client.py:514). The dlopen runs the plugin's SCA_REGISTER_*
  // constructors against this module's registries.
  m.def("load_op_library", [](const std::string& path) {
    void* h = dlopen(path.c_str(), RTLD_NOW | RTLD_LOCAL);
    if (!h) {
      throw ScannerError(std::string("load_op_library failed: ") +
                         dlerror());
    }
  });

  m.def("have_gpu", &have_gpu);
  m.def("gpu_device_count", &gpu_device_count);
  m.def("gpu_free_memory", [](int device) -> size_t {
    SCA_CHECK(have_gpu(), "gpu_free_memory needs a GPU");
    int prev = 0;
    (void)hipGetDevice(&prev);
    (void)hipSetDevice(device);
    size_t free_b = 0, total_b = 0;
    hipError_t e = hipMemGetInfo(&free_b, &total_b);
    (void)hipSetDevice(prev);
    SCA_CHECK(e == hipSuccess, "hipMemGetInfo failed");
    return free_b;
  });

  // GEMM micro-benchmark: device-resident buffers, hipEvent timing.
  // Returns ms per iteration (tools/gemm_tune.py computes TFLOP/s).
  m.def("gemm_bench", [](int M, int N, int K, int iters, bool relu) {
    SCA_CHECK(have_gpu(), "gemm_bench needs a GPU");
    DeviceHandle dev{DeviceType::GPU, 0};
    u8* dA = new_buffer(dev, (size_t)M * K * 2);
    u8* dB = new_buffer(dev, (size_t)N * K * 2);
    u8* dC = new_buffer(dev, (size_t)M * N * 2);
    u8* dS = new_buffer(dev, (size_t)N * 8);
    u8* dSK = new_buffer(dev, 64u << 20);
    splitk_scratch_init(dSK, 64u << 20, nullptr);
    GemmArgs g;
    g.splitk_scratch = dSK;
    g.splitk_scratch_bytes = 64u << 20;
    g.A = dA;
    g.B = dB;
    g.C = dC;
    g.M = M;
    g.N = N;
    g.K = K;
    g.scale = (const float*)dS;
    g.bias = (const float*)(dS + (size_t)N * 4);
    g.relu = relu;
    void* s = per_thread_hip_stream();
    for (int i = 0; i < 3; ++i) gemm_bf16(g, s);
    sync_per_thread_stream();
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    (void)hipEventRecord(e0, (hipStream_t)s);
    for (int i = 0; i < iters; ++i) gemm_bf16(g, s);
    (void)hipEventRecord(e1, (hipStream_t)s);
    sync_per_thread_stream();
    float ms = 0.f;
    (void)hipEventElapsedTime(&ms, e0, e1);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    delete_buffer(dev, dA);
    delete_buffer(dev, dB);
    delete_buffer(dev, dC);
    delete_buffer(dev, dS);
    delete_buffer(dev, dSK);
    return ms / iters;
  });

  // Raw MFMA GEMM entry for numerics tests: f32 inputs are rounded to
  // bf16 on the host, C = A @ B_nk^T computed on GPU 0.
  m.def("gemm_bf16_test",
        [](py::array_t<float, py::array::c_style | py::array::forcecast> A,
           py::array_t<float, py::array::c_style | py::array::forcecast> Bnk,
           bool relu) {
          SCA_CHECK(have_gpu(), "gemm_bf16_test needs a GPU");
          int M = (int)A.shape(0), K = (int)A.shape(1);
          int N = (int)Bnk.shape(0);
          SCA_CHECK((int)Bnk.shape(1) == K, "K mismatch");
          DeviceHandle dev{DeviceType::GPU, 0};
          auto to_bf16 = [](const float* p, size_t n) {
            std::vector<u16> v(n);
            for (size_t i = 0; i < n; ++i) {
              u32 bits;
              std::memcpy(&bits, &p[i], 4);
              bits += 0x7fff + ((bits >> 16) & 1);
              v[i] = (u16)(bits >> 16);
            }
            return v;
          };
          auto ah = to_bf16(A.data(), (size_t)M * K);
          auto bh = to_bf16(Bnk.data(), (size_t)N * K);
          u8* dA = new_buffer(dev, ah.size() * 2);
          u8* dB = new_buffer(dev, bh.size() * 2);
          u8* dC = new_buffer(dev, (size_t)M * N * 2);
          memcpy_buffer(dA, dev, (u8*)ah.data(), CPU_DEVICE, ah.size() * 2);
          memcpy_buffer(dB, dev, (u8*)bh.data(), CPU_DEVICE, bh.size() * 2);
          GemmArgs g;
          g.A = dA;
          g.B = dB;
          g.C = dC;
          g.M = M;
          g.N = N;
          g.K = K;
          g.relu = relu;
          gemm_bf16(g, per_thread_hip_stream());
          sync_per_thread_stream();
          std::vector<u16> ch((size_t)M * N);
          memcpy_buffer((u8*)ch.data(), CPU_DEVICE, dC, dev, ch.size() * 2);
          delete_buffer(dev, dA);
          delete_buffer(dev, dB);
          delete_buffer(dev, dC);
          py::array_t<float> out({M, N});
          float* op = out.mutable_data();
          for (size_t i = 0; i < ch.size(); ++i) {
            u32 bits = (u32)ch[i] << 16;
            std::memcpy(&op[i], &bits, 4);
          }
          return out;
        });

  m.def("init_memory", [](size_t cpu_pool, size_t gpu_pool,
                          std::vector<i32> gpu_ids) {
    MemoryConfig cfg;
    cfg.cpu_pool_size = cpu_pool;
    cfg.gpu_pool_size = gpu_pool;
    cfg.gpu_ids = std::move(gpu_ids);
    init_memory_allocators(cfg);
  });
  m.def("destroy_memory", &destroy_memory_allocators);
  // HBM span-cache introspection (tests + profiling)
  m.def("span_cache_stats", [] {
    py::dict d;
    d["hits"] = span_cache_hits();
    d["misses"] = span_cache_misses();
    d["bytes"] = span_cache_bytes_live();
    d["budget"] = span_cache_budget();
    return d;
  });
  // kernel-isolation test: CPU-encode -> GPU-decode -> bytes back (no
  // engine, no cache, no chunking)
  m.def("svc_gpu_debug",
        [](py::array_t<u8, py::array::c_style | py::array::forcecast> frames) {
          i64 n = frames.shape(0);
          i32 h = (i32)frames.shape(1), w = (i32)frames.shape(2),
              c = (i32)frames.shape(3);
          VideoMetadata vm;
          std::vector<u8> stream;
          svc_encode_cpu(frames.data(), n, h, w, c, 16, stream, vm);
          py::dict r;
          r["dev"] = svc_gpu_debug_dump(stream, vm);
          // host-side expectations for supers 255..257, frame 0
          u32 nbytes = (u32)((i64)h) * w * c;
          u32 ngroups = (nbytes + 31) / 32;
          u32 nsuper = (ngroups + 127) / 128;
          const u8* pkt = stream.data() + vm.sample_offsets[0];
          const u32* so = (const u32*)(pkt + 20);
          u64 widths_off = 20ull + (u64)nsuper * 4;
          u64 packed_off = widths_off + (ngroups + 3) / 4 * 4;
          py::list host;
          for (u32 s = 255; s <= 257; ++s) {
            u32 g0 = s * 128;
            u32 wv = pkt[widths_off + g0];
            const u32* q = (const u32*)(pkt + packed_off + so[s]);
            py::dict e;
            e["super_off"] = so[s];
            e["w_lane0"] = wv;
            e["q0"] = q[0];
            e["q1"] = q[1];
            e["packed_off"] = (u32)packed_off;
            host.append(e);
          }
          r["host"] = host;
          return r;
        });
  m.def("svc_gpu_roundtrip",
        [](py::array_t<u8, py::array::c_style | py::array::forcecast> frames,
           i32 gop, std::vector<i64> want) {
          SCA_CHECK(frames.ndim() == 4, "frames must be [N,H,W,C]");
          i64 n = frames.shape(0);
          i32 h = (i32)frames.shape(1), w = (i32)frames.shape(2),
              c = (i32)frames.shape(3);
          VideoMetadata vm;
          std::vector<u8> stream;
          svc_encode_cpu(frames.data(), n, h, w, c, gop, stream, vm);
          if (want.empty())
            for (i64 i = 0; i < n; ++i) want.push_back(i);
          DeviceHandle dev{DeviceType::GPU, 0};
          auto elems = svc_decode_gpu(stream.data(), stream.size(), vm,
                                      want, dev, 0);
          py::array_t<u8> out({(i64)elems.size(), (i64)h, (i64)w, (i64)c});
          size_t fsize = (size_t)h * w * c;
          for (size_t i = 0; i < elems.size(); ++i) {
            memcpy_buffer(out.mutable_data() + i * fsize, CPU_DEVICE,
                          elems[i].buffer, dev, fsize);
            delete_buffer(dev, elems[i].buffer);
          }
          return out;
        },
        py::arg("frames"), py::arg("gop") = 16,
        py::arg("want") = std::vector<i64>{});
  m.def("span_cache_clear", &span_cache_clear);
  m.def("span_cache_set_budget", &span_cache_set_budget);
  // engine memory accounting (device -1 = CPU)
  m.def("mem_stats", [](int device) {
    DeviceHandle d = device < 0 ? CPU_DEVICE
                                : DeviceHandle{DeviceType::GPU, device};
    py::dict r;
    r["live"] = mem_bytes_live(d);
    r["peak"] = mem_bytes_peak(d);
    return r;
  });
  m.def("mem_reset_peak", [](int device) {
    DeviceHandle d = device < 0 ? CPU_DEVICE
                                : DeviceHandle{DeviceType::GPU, device};
    mem_reset_peak(d);
  });

  m.def("registered_ops", [] { return op_registry().names(); });
  m.def("registered_sources", [] { return source_registry().names(); });
  m.def("registered_sinks", [] { return sink_registry().names(); });
  m.def("op_info", [](const std::string& name) {
    const OpInfo& o = op_registry().get(name);
    py::dict d;
    py::list ins, outs;
    for (auto& c : o.input_columns)
      ins.append(py::make_tuple(c.name, (i32)c.type));
    for (auto& c : o.output_columns)
      outs.append(py::make_tuple(c.name, (i32)c.type));
    d["input_columns"] = ins;
    d["output_columns"] = outs;
    d["variadic"] = o.variadic_inputs;
    d["stencil"] = o.stencil;
    d["bounded_state"] = o.has_bounded_state;
    d["warmup"] = o.warmup;
    d["unbounded_state"] = o.has_unbounded_state;
    return d;
  });
  m.def("has_kernel", [](const std::string& op, i32 device) {
    return kernel_registry().has(op, (DeviceType)device);
  });
  m.def("register_python_op", &register_python_op_binding);

  py::class_<Database, std::shared_ptr<Database>>(m, "Database")
      .def(py::init([](const std::string& path) {
        return std::make_shared<Database>(StorageBackend::make_posix(), path);
      }))
      .def(py::init([](const std::string& path,
                       const std::string& storage_type,
                       const std::string& bucket) {
        std::unique_ptr<StorageBackend> s;
        if (storage_type == "posix") {
          s = StorageBackend::make_posix();
        } else if (storage_type == "s3" || storage_type == "gcs" ||
                   storage_type == "object") {
          SCA_CHECK(!bucket.empty(),
                    "object storage needs a bucket path");
          s = StorageBackend::make_object_store(bucket);
        } else {
          throw ScannerError("unknown storage type '" + storage_type + "'");
        }
        return std::make_shared<Database>(std::move(s), path);
      }))
      .def("recover", &Database::recover)
      .def("write_megafile", &Database::write_megafile)
      .def("table_names", &Database::table_names)
      .def("has_table", &Database::has_table)
      .def("delete_table", &Database::delete_table)
      .def("table_committed",
           [](Database& db, const std::string& name) {
             return db.table_committed(db.get_table(name).id);
           })
      .def("table_info", [](Database& db, const std::string& name) {
        TableMetadata t = db.get_table(name);
        py::dict d;
        d["id"] = t.id;
        d["name"] = t.name;
        d["num_rows"] = t.num_rows();
        py::list cols;
        for (auto& c : t.columns)
          cols.append(py::make_tuple(c.name, (i32)c.type));
        d["columns"] = cols;
        d["end_rows"] = t.end_rows;
        d["committed"] = db.table_committed(t.id);
        return d;
      });

  // ---- table write/read helpers (client-side ingest + load) ----

  m.def("write_bytes_table",
        [](std::shared_ptr<Database> db, const std::string& name,
           const std::vector<std::string>& columns,
           const std::vector<std::vector<py::bytes>>& rows_per_column,
           i64 io_packet_size) {
          std::vector<ColumnType> types(columns.size(), ColumnType::Bytes);
          TableMetadata t = db->new_table(name, columns, types, true);
          i64 n = rows_per_column.empty() ? 0 : (i64)rows_per_column[0].size();
          std::vector<i64> ends;
          i32 item = 0;
          for (i64 s = 0; s < n; s += io_packet_size, ++item) {
            i64 e = std::min(n, s + io_packet_size);
            for (size_t c = 0; c < columns.size(); ++c) {
              std::vector<Element> elems;
              std::vector<std::string> bufs;
              for (i64 r = s; r < e; ++r) {
                bufs.push_back(rows_per_column[c][r]);
              }
              for (auto& bstr : bufs) {
                Element el;
                el.buffer = reinterpret_cast<u8*>(bstr.data());
                el.size = bstr.size();
                elems.push_back(el);
              }
              write_column_item(*db, t, columns[c], item, elems);
            }
            ends.push_back(e);
          }
          if (ends.empty()) ends.push_back(0);
          t.end_rows = ends;
          db->update_table(t);
          db->commit_table(t.id);
        });

  // ---- real-video ingest (mp4/Annex-B -> indexed h264 table) ----
  m.def("ingest_video_file",
        [](std::shared_ptr<Database> db, const std::string& name,
           const std::string& column, const std::string& path) {
          IngestResult r = ingest_video_file(*db, name, column, path);
          py::dict d;
          d["num_frames"] = r.num_frames;
          d["width"] = r.width;
          d["height"] = r.height;
          d["codec"] = r.codec;
          return d;
        });
  m.def("export_mp4",
        [](std::shared_ptr<Database> db, const std::string& name,
           const std::string& column, const std::string& path, double fps) {
          export_mp4(*db, name, column, path, fps);
        });
  // parser introspection (unit tests)
  m.def("h264_index", [](py::bytes b) {
    std::string s = b;
    H264Index idx = h264_index_annexb((const u8*)s.data(), s.size());
    py::dict d;
    d["width"] = idx.width;
    d["height"] = idx.height;
    d["num_frames"] = idx.num_frames;
    d["sample_offsets"] = idx.sample_offsets;
    d["sample_sizes"] = idx.sample_sizes;
    d["keyframe_indices"] = idx.keyframe_indices;
    d["sps"] = py::bytes((const char*)idx.sps.data(), idx.sps.size());
    d["pps"] = py::bytes((const char*)idx.pps.data(), idx.pps.size());
    return d;
  });
  m.def("mp4_probe", [](py::bytes b) {
    std::string s = b;
    Mp4Track t = mp4_parse((const u8*)s.data(), s.size());
    py::dict d;
    d["width"] = t.width;
    d["height"] = t.height;
    d["length_size"] = t.length_size;
    d["sample_offsets"] = t.sample_offsets;
    d["sample_sizes"] = t.sample_sizes;
    d["keyframe_indices"] = t.keyframe_indices;
    d["timescale"] = t.timescale;
    d["n_sps"] = (i64)t.sps.size();
    d["n_pps"] = (i64)t.pps.size();
    return d;
  });
  m.def("h264_parse_sps_py", [](py::bytes b) {
    std::string s = b;
    H264Sps sps = h264_parse_sps((const u8*)s.data(), s.size());
    py::dict d;
    d["width"] = sps.width;
    d["height"] = sps.height;
    d["profile_idc"] = sps.profile_idc;
    return d;
  });

  m.def("write_video_table",
        [](std::shared_ptr<Database> db, const std::string& name,
           const std::string& column,
           py::array_t<u8, py::array::c_style | py::array::forcecast> frames,
           i64 io_packet_size, const std::string& codec) {
          SCA_CHECK(frames.ndim() == 4, "frames must be [N,H,W,C] u8");
          i64 n = frames.shape(0);
          i32 h = (i32)frames.shape(1), w = (i32)frames.shape(2),
              c = (i32)frames.shape(3);
          size_t fsize = (size_t)h * w * c;
          TableMetadata t =
              db->new_table(name, {column}, {ColumnType::Video}, true);
          const u8* data = frames.data();
          std::vector<i64> ends;
          i32 item = 0;
          for (i64 s = 0; s < n; s += io_packet_size, ++item) {
            i64 e = std::min(n, s + io_packet_size);
            VideoMetadata vm;
            vm.width = w;
            vm.height = h;
            vm.channels = c;
            vm.frame_type = FrameType::U8;
            vm.num_frames = e - s;
            if (codec == "raw") {
              vm.codec = "raw";
              std::vector<Element> elems;
              for (i64 r = s; r < e; ++r) {
                Element el;
                el.buffer = const_cast<u8*>(data + (size_t)r * fsize);
                el.size = fsize;
                elems.push_back(el);
              }
              write_column_item(*db, t, column, item, elems);
              for (i64 r = s; r < e; ++r) {
                vm.keyframe_indices.push_back(r - s);
                vm.sample_offsets.push_back((r - s) * fsize);
                vm.sample_sizes.push_back(fsize);
              }
              auto vbuf = vm.serialize();
              db->storage()->write_all(
                  db->paths().video_metadata(t.id, t.column_id(column), item),
                  vbuf.data(), vbuf.size());
            } else if (codec == "svc") {
              std::vector<u8> stream;
              svc_encode_cpu(data + (size_t)s * fsize, e - s, h, w, c,
                             /*gop=*/16, stream, vm);
              write_video_item(*db, t, column, item, stream, vm);
            } else {
              throw ScannerError("unknown codec '" + codec + "'");
            }
            ends.push_back(e);
          }
          if (ends.empty()) ends.push_back(0);
          t.end_rows = ends;
          db->update_table(t);
          db->commit_table(t.id);
        },
        py::arg("db"), py::arg("name"), py::arg("column"), py::arg("frames"),
        py::arg("io_packet_size") = 128, py::arg("codec") = "raw");

  m.def("read_column",
        [](std::shared_ptr<Database> db, const std::string& table,
           const std::string& column, std::vector<i64> rows) {
          TableMetadata t = db->get_table(table);
          if (rows.empty()) {
            for (i64 r = 0; r < t.num_rows(); ++r) rows.push_back(r);
          }
          py::list out;
          if (t.column_type(column) == ColumnType::Video) {
            auto items = items_for_rows(t, rows);
            size_t ri = 0;
            for (auto& ir : items) {
              VideoMetadata vm = read_video_metadata(*db, t, column, ir.item);
              std::vector<i64> local_rows;
              while (ri < rows.size() && rows[ri] < ir.row_end) {
                local_rows.push_back(rows[ri]);
                ++ri;
              }
              if (vm.codec == "raw") {
                ElementVector elems =
                    read_column_rows(*db, t, column, local_rows, 8);
                for (auto& e : elems) {
                  py::tuple shape =
                      py::make_tuple(vm.height, vm.width, vm.channels);
                  out.append(py::make_tuple(
                      py::bytes((const char*)e.buffer, e.size), shape,
                      (i32)vm.frame_type));
                  delete_buffer(CPU_DEVICE, e.buffer);
                }
              } else {
                // decode requested frames on CPU
                std::vector<i64> want;
                for (i64 r : local_rows) want.push_back(r - ir.row_start);
                auto stream = db->storage()->read_all(
                    db->paths().item(t.id, t.column_id(column), ir.item));
                std::vector<std::vector<u8>> frames;
                svc_decode_cpu(stream.data(), stream.size(), vm, want, frames);
                for (auto& fb : frames) {
                  py::tuple shape =
                      py::make_tuple(vm.height, vm.width, vm.channels);
                  out.append(py::make_tuple(
                      py::bytes((const char*)fb.data(), fb.size()), shape,
                      (i32)vm.frame_type));
                }
              }
            }
          } else {
            ElementVector elems = read_column_rows(*db, t, column, rows, 8);
            for (auto& e : elems) {
              if (e.size == 0) {
                out.append(py::none());
              } else {
                out.append(py::bytes((const char*)e.buffer, e.size));
              }
              if (e.buffer) delete_buffer(CPU_DEVICE, e.buffer);
            }
          }
          return out;
        });

  // ---- analysis (exposed for unit tests) ----

  m.def("analyze_job_py", [](const py::bytes& graph_b, const py::bytes& job_b,
                             const py::dict& table_rows) {
    JobGraph g = graph_from_bytes(graph_b);
    validate_graph(g);
    JobBinding job = JobBinding::from_msgpack(mp_from_pybytes(job_b));
    std::map<std::string, i64> rows_map;
    for (auto item : table_rows)
      rows_map[item.first.cast<std::string>()] = item.second.cast<i64>();
    auto ja = analyze_job(g, job, [&](const SourceArgsC& s) {
      return rows_map.at(s.table);
    });
    py::dict d;
    py::list doms;
    for (auto& dom : ja.domains) {
      py::dict dd;
      dd["num_rows"] = dom.num_rows;
      dd["slice_level"] = dom.slice_level;
      dd["group_starts"] = dom.group_starts;
      doms.append(dd);
    }
    d["domains"] = doms;
    d["output_rows"] = ja.output_rows;
    return d;
  });

  m.def("task_plan_py", [](const py::bytes& graph_b, const py::bytes& job_b,
                           const py::dict& table_rows, i64 start, i64 end) {
    JobGraph g = graph_from_bytes(graph_b);
    validate_graph(g);
    JobBinding job = JobBinding::from_msgpack(mp_from_pybytes(job_b));
    std::map<std::string, i64> rows_map;
    for (auto item : table_rows)
      rows_map[item.first.cast<std::string>()] = item.second.cast<i64>();
    auto ja = analyze_job(g, job, [&](const SourceArgsC& s) {
      return rows_map.at(s.table);
    });
    auto plan = derive_task_plan(g, ja, job, start, end);
    py::dict d;
    py::list ops;
    for (auto& otp : plan.ops) {
      py::dict od;
      od["required_rows"] = otp.required_rows;
      od["compute_rows"] = otp.compute_rows;
      od["windows"] = otp.windows;
      od["remap"] = otp.remap;
      py::list rb;
      for (u8 v : otp.reset_before) rb.append((bool)v);
      od["reset_before"] = rb;
      ops.append(od);
    }
    d["ops"] = ops;
    py::dict lr;
    for (auto& kv : plan.load_rows) lr[py::int_(kv.first)] = kv.second;
    d["load_rows"] = lr;
    return d;
  });

  // ---- execution ----

  py::class_<PreparedTask, std::shared_ptr<PreparedTask>>(m, "PreparedTask");

  py::class_<LocalExecutor>(m, "LocalExecutor")
      .def(py::init([](std::shared_ptr<Database> db, const py::bytes& graph_b,
                       const py::bytes& jobs_b, const py::dict& perf,
                       std::vector<i32> gpu_ids) {
             return new LocalExecutor(db, graph_from_bytes(graph_b),
                                      jobs_from_bytes(jobs_b),
                                      perf_from_dict(perf), gpu_ids);
           }),
           py::arg("db"), py::arg("graph"), py::arg("jobs"), py::arg("perf"),
           py::arg("gpu_ids") = std::vector<i32>{})
      .def("run",
           [](LocalExecutor& ex) {
             py::gil_scoped_release rel;
             ex.run();
           })
      .def("prepare", &LocalExecutor::prepare,
           py::arg("create_outputs") = true)
      .def("all_tasks",
           [](LocalExecutor& ex) {
             py::list out;
             for (auto& t : ex.all_tasks())
               out.append(py::make_tuple(t.job, t.task, t.start, t.end));
             return out;
           })
      .def("process_task",
           [](LocalExecutor& ex, i32 instance, i32 job, i32 task, i64 start,
              i64 end) {
             TaskDesc t{job, task, start, end};
             py::gil_scoped_release rel;
             ex.process_task_public(instance, t);
           })
      .def("prepare_task",
           [](LocalExecutor& ex, i32 job, i32 task, i64 start, i64 end) {
             TaskDesc t{job, task, start, end};
             py::gil_scoped_release rel;
             return ex.prepare_task_public(t);
           })
      .def("process_prepared",
           [](LocalExecutor& ex, i32 instance,
              std::shared_ptr<PreparedTask> pt) {
             py::gil_scoped_release rel;
             ex.process_prepared_public(instance, pt);
           })
      .def("finalize_job", &LocalExecutor::finalize_job)
      .def("tasks_done", &LocalExecutor::tasks_done)
      .def("total_output_rows", &LocalExecutor::total_output_rows)
      .def("profilers",
           [](LocalExecutor& ex) { return profilers_to_py(ex.profilers()); });
}
