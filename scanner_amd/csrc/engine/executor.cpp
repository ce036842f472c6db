#include "executor.h"

#include <algorithm>
#include <set>

#include "../hip_util.h"
#include "../memory.h"
#include "../ops/source.h"
#include "../video/svc.h"

namespace sca {

Element element_to_device(const Element& e, DeviceHandle target, bool& copied) {
  copied = false;
  if (e.is_null || e.device == target) return e;
  Element out = e;
  out.device = target;
  out.buffer = new_buffer(target, e.size);
  memcpy_buffer(out.buffer, target, e.buffer, e.device, e.size);
  copied = true;
  return out;
}

struct LocalExecutor::Instance {
  i32 idx = 0;
  DeviceHandle gpu{DeviceType::CPU, 0};  // GPU assigned to this instance
  Profiler* profiler = nullptr;
  // kernel instances per op (null for builtins); re-created per job when
  // per-stream args change
  std::vector<std::unique_ptr<BaseKernel>> kernels;
  i32 kernels_job = -1;
};

LocalExecutor::LocalExecutor(std::shared_ptr<Database> db, JobGraph graph,
                             std::vector<JobBinding> jobs, PerfParams pp,
                             std::vector<i32> gpu_ids)
    : db_(std::move(db)),
      graph_(std::move(graph)),
      jobs_(std::move(jobs)),
      pp_(pp),
      gpu_ids_(std::move(gpu_ids)) {
  MemoryConfig mc;
  mc.cpu_pool_size = pp_.cpu_pool_size;
  mc.gpu_pool_size = pp_.gpu_pool_size;
  mc.gpu_ids = gpu_ids_;
  init_memory_allocators(mc);
  db_key_ = std::hash<std::string>{}(db_->paths().root);
  if (!gpu_ids_.empty()) {
    size_t budget = pp_.span_cache_size;
    if (budget == 0) {
      budget = pp_.gpu_pool_size
                   ? std::min<size_t>(pp_.gpu_pool_size / 3, 16ull << 30)
                   : (4ull << 30);
    }
    span_cache_set_budget(budget);
  }
}

LocalExecutor::~LocalExecutor() = default;

void LocalExecutor::make_instance(i32 idx) {
  auto inst = std::make_unique<Instance>();
  inst->idx = idx;
  if (!gpu_ids_.empty()) {
    inst->gpu = DeviceHandle{DeviceType::GPU,
                             gpu_ids_[idx % gpu_ids_.size()]};
  }
  profilers_.push_back(std::make_unique<Profiler>(
      (ProfilerLevel)pp_.profiler_level));
  inst->profiler = profilers_.back().get();
  instances_.push_back(std::move(inst));
}

void LocalExecutor::prepare(bool create_outputs) {
  if (prepared_) return;
  validate_graph(graph_);
  SourceRowsFn source_rows = [this](const SourceArgsC& s) -> i64 {
    if (!s.source.empty()) {
      // user source: the registered enumerator names the row domain
      SourceConfig cfg;
      cfg.args = s.args;
      return source_registry().get(s.source).make_enumerator(cfg)
          ->total_elements();
    }
    return db_->get_table(s.table).num_rows();
  };
  for (auto& job : jobs_) {
    analyses_.push_back(analyze_job(graph_, job, source_rows));
  }
  // Create output tables (uncommitted), one item per task.
  const OpNode& out_op = graph_.ops.back();
  for (size_t j = 0; j < jobs_.size(); ++j) {
    std::vector<std::string> col_names;
    std::vector<ColumnType> col_types;
    for (auto& e : out_op.inputs) {
      col_names.push_back(e.column);
      const auto& psi = analyses_[j].info[e.op];
      size_t ci = 0;
      for (; ci < psi.output_columns.size(); ++ci)
        if (psi.output_columns[ci] == e.column) break;
      SCA_CHECK(ci < psi.output_columns.size(),
                "Output references unknown column '" + e.column + "'");
      col_types.push_back(psi.output_types[ci]);
    }
    TableMetadata t;
    if (jobs_[j].sink.sink.empty()) {
      t = create_outputs
              ? db_->new_table(jobs_[j].sink.table, col_names, col_types,
                               /*overwrite=*/true)
              : db_->get_table(jobs_[j].sink.table);
    }  // user sink: no output table
    out_tables_.push_back(t);
    // task boundaries
    i64 rows = analyses_[j].output_rows;
    std::vector<i64> ends;
    for (i64 s = 0; s < rows; s += pp_.io_packet_size) {
      ends.push_back(std::min(rows, s + pp_.io_packet_size));
    }
    if (ends.empty()) ends.push_back(0);
    task_rows_.push_back(ends);
  }
  for (i32 i = 0; i < pp_.pipeline_instances; ++i) make_instance(i);
  prepared_ = true;
}

std::vector<TaskDesc> LocalExecutor::all_tasks() const {
  std::vector<TaskDesc> tasks;
  for (size_t j = 0; j < task_rows_.size(); ++j) {
    i64 prev = 0;
    for (size_t t = 0; t < task_rows_[j].size(); ++t) {
      tasks.push_back(TaskDesc{(i32)j, (i32)t, prev, task_rows_[j][t]});
      prev = task_rows_[j][t];
    }
  }
  return tasks;
}

i64 LocalExecutor::total_output_rows() const {
  i64 total = 0;
  for (auto& a : analyses_) total += a.output_rows;
  return total;
}

void LocalExecutor::finalize_job(i32 job) {
  if (!jobs_[job].sink.sink.empty()) return;  // user sink: nothing to commit
  TableMetadata t = out_tables_[job];
  t.end_rows = task_rows_[job];
  db_->update_table(t);
  db_->commit_table(t.id);
}

void LocalExecutor::run() {
  prepare();
  auto tasks = all_tasks();
  BoundedQueue<TaskDesc> q_raw;
  for (auto& t : tasks) q_raw.push(t);
  q_raw.close();

  std::atomic<bool> failed{false};
  std::string fail_msg;
  std::mutex fail_mu;
  auto record_failure = [&](const std::exception& ex) {
    std::lock_guard<std::mutex> l(fail_mu);
    if (!failed.exchange(true)) fail_msg = ex.what();
  };

  // Load-worker stage (reference: dedicated load threads feeding the
  // pipeline, worker.cpp:85): derive plans + read video source spans
  // ahead of the instances, bounded queue for backpressure so at most a
  // few tasks' pinned bytes are in flight.
  i32 n_load = pp_.load_workers > 0
                   ? pp_.load_workers
                   : std::min<i32>(8, std::max<i32>(2, (i32)instances_.size()));
  BoundedQueue<std::shared_ptr<PreparedTask>> q_ready(
      std::max<size_t>((size_t)pp_.queue_size, instances_.size() + 2));
  std::vector<std::unique_ptr<Profiler>> load_profilers;
  for (i32 i = 0; i < n_load; ++i) {
    load_profilers.push_back(
        std::make_unique<Profiler>((ProfilerLevel)pp_.profiler_level));
  }
  std::atomic<i32> live_loaders{n_load};
  auto loader = [&](i32 li) {
    while (auto t = q_raw.try_pop()) {
      if (failed.load()) break;
      try {
        q_ready.push(prepare_task(*t, load_profilers[li].get()));
      } catch (const std::exception& ex) {
        record_failure(ex);
        break;
      }
    }
    if (live_loaders.fetch_sub(1) == 1) q_ready.close();
  };

  auto worker = [&](i32 idx) {
    Instance& inst = *instances_[idx];
    while (auto pt = q_ready.pop()) {
      if (failed.load()) return;
      try {
        process_task(inst, (*pt)->desc, pt->get());
      } catch (const std::exception& ex) {
        record_failure(ex);
        // Unblock loaders waiting on a full ready-queue.
        q_ready.close();
        return;
      }
    }
  };

  std::vector<std::thread> threads;
  for (i32 i = 0; i < n_load; ++i) threads.emplace_back(loader, i);
  for (size_t i = 0; i < instances_.size(); ++i)
    threads.emplace_back(worker, (i32)i);
  for (auto& th : threads) th.join();
  // Failure paths can leave prepared tasks queued; drain so their pinned
  // buffers return to the pool before profilers move.
  while (auto leftover = q_ready.try_pop()) leftover->reset();
  for (auto& lp : load_profilers) profilers_.push_back(std::move(lp));
  if (failed.load()) throw ScannerError("job failed: " + fail_msg);
  for (size_t j = 0; j < jobs_.size(); ++j) finalize_job((i32)j);
}

void LocalExecutor::process_task_public(i32 instance, const TaskDesc& t) {
  prepare();
  while ((i32)instances_.size() <= instance) make_instance((i32)instances_.size());
  process_task(*instances_[instance], t);
}

std::shared_ptr<PreparedTask> LocalExecutor::prepare_task_public(
    const TaskDesc& t) {
  prepare();
  static thread_local Profiler loader_prof(ProfilerLevel::Info);
  return prepare_task(t, &loader_prof);
}

void LocalExecutor::process_prepared_public(
    i32 instance, const std::shared_ptr<PreparedTask>& pt) {
  prepare();
  while ((i32)instances_.size() <= instance)
    make_instance((i32)instances_.size());
  process_task(*instances_[instance], pt->desc, pt.get());
}

PreparedTask::~PreparedTask() {
  for (auto& sp : spans) {
    if (sp.buf) delete_buffer(CPU_DEVICE, sp.buf);
  }
}

u8* PreparedTask::take_span(i32 op, i32 item, u64 lo, u64 hi) {
  for (auto& sp : spans) {
    if (sp.buf && sp.op == op && sp.item == item && sp.lo == lo &&
        sp.hi == hi) {
      u8* b = sp.buf;
      sp.buf = nullptr;  // ownership moves to the caller
      return b;
    }
  }
  return nullptr;
}

SpanHandle PreparedTask::take_cached(i32 op, i32 item, u64 lo, u64 hi) {
  for (auto& sp : spans) {
    if (sp.cached && sp.op == op && sp.item == item && sp.lo == lo &&
        sp.hi == hi) {
      return std::move(sp.cached);
    }
  }
  return nullptr;
}

SpanHandle LocalExecutor::acquire_ready_span(const TableMetadata& table,
                                             const std::string& column,
                                             i32 item, u64 lo, u64 hi,
                                             DeviceHandle gpu, Profiler* prof,
                                             const u8* host_bytes) {
  SpanKey k;
  k.db = db_key_;
  k.table = table.id;
  k.col = table.column_id(column);
  k.item = item;
  k.lo = lo;
  k.hi = hi;
  auto res = span_cache_acquire(gpu, k, hi - lo);
  if (!res.first) return nullptr;
  if (!res.second) {
    // someone else owns the upload; wait for it (uploads are ms-scale)
    if (!span_wait_ready(res.first)) return nullptr;
    return res.first;
  }
  u8* tmp = nullptr;
  try {
    const u8* src_bytes = host_bytes;
    if (!src_bytes) {
      tmp = new_buffer(CPU_DEVICE, hi - lo);
      Profiler::Scope sl(prof, "load:video");
      db_->storage()->read_range(db_->paths().item(table.id, k.col, item),
                                 lo, hi - lo, tmp);
      prof->increment("io_read_bytes", (i64)(hi - lo));
      src_bytes = tmp;
    }
    Profiler::Scope su(prof, "h2d:span");
    span_cache_upload(res.first, src_bytes, hi - lo);
  } catch (...) {
    span_mark_failed(res.first);
    if (tmp) delete_buffer(CPU_DEVICE, tmp);
    throw;
  }
  if (tmp) delete_buffer(CPU_DEVICE, tmp);
  return res.first;
}

// Load-worker half of a task (runs on the prefetch pool): derive the plan
// and read every svc video source span into pinned memory so the pipeline
// instance starts with its input bytes already resident.
std::shared_ptr<PreparedTask> LocalExecutor::prepare_task(const TaskDesc& t,
                                                          Profiler* prof) {
  auto pt = std::make_shared<PreparedTask>();
  pt->desc = t;
  const JobBinding& job = jobs_[t.job];
  {
    Profiler::Scope s(prof, "derive_task_plan");
    pt->plan = derive_task_plan(graph_, analyses_[t.job], job, t.start,
                                t.end);
  }
  for (auto& kv : pt->plan.load_rows) {
    i32 op_idx = kv.first;
    const std::vector<i64>& rows = kv.second;
    const SourceArgsC& src = job.sources.at(op_idx);
    const OpStaticInfo& si = analyses_[t.job].info[op_idx];
    if (si.output_types[0] != ColumnType::Video) continue;
    TableMetadata table = db_->get_table(src.table);
    auto items = items_for_rows(table, rows);
    size_t ri = 0;
    for (auto& ir : items) {
      VideoMetadata m = read_video_metadata(*db_, table, src.column,
                                            ir.item);
      std::vector<i64> local;
      while (ri < rows.size() && rows[ri] < ir.row_end) {
        local.push_back(rows[ri] - ir.row_start);
        ++ri;
      }
      if (m.codec != "svc") continue;
      std::vector<i64> span = svc_decode_span(m, local);
      if (span.empty()) continue;
      u64 lo = m.sample_offsets[span.front()];
      u64 hi = m.sample_offsets[span.back()] + m.sample_sizes[span.back()];
      PreparedTask::Span sp;
      sp.op = op_idx;
      sp.item = ir.item;
      sp.lo = lo;
      sp.hi = hi;
      // HBM span cache first: a hit skips both the storage read and the
      // H2D; a fresh insert does read+upload HERE on the load worker, off
      // the pipeline instances' critical path (overlap
      // IO/H2D with GPU compute).
      if (!gpu_ids_.empty()) {
        DeviceHandle tgt{DeviceType::GPU,
                         gpu_ids_[t.task % (i32)gpu_ids_.size()]};
        sp.cached = acquire_ready_span(table, src.column, ir.item, lo, hi,
                                       tgt, prof, nullptr);
      }
      if (!sp.cached) {
        sp.buf = new_buffer(CPU_DEVICE, hi - lo);
        {
          Profiler::Scope sl(prof, "load:video");
          db_->storage()->read_range(
              db_->paths().item(table.id, table.column_id(src.column),
                                ir.item),
              lo, hi - lo, sp.buf);
        }
        prof->increment("io_read_bytes", (i64)(hi - lo));
      }
      pt->spans.push_back(sp);
    }
  }
  return pt;
}

// ---- source loading -------------------------------------------------
// Non-video and raw-codec columns load eagerly (cheap byte reads); svc
// video sources get an incremental decode state per item so the streaming
// packet loop pulls frames as needed instead of materializing the whole
// task's decode output up front.
void LocalExecutor::init_sources(
    Instance& inst, const TaskDesc& t, const TaskPlan& plan, OutsVec& outs,
    std::map<i32, std::vector<SvcItemState>>& svc_states, PreparedTask* prep) {
  const JobBinding& job = jobs_[t.job];
  for (auto& kv : plan.load_rows) {
    i32 op_idx = kv.first;
    const std::vector<i64>& rows = kv.second;
    const SourceArgsC& src = job.sources.at(op_idx);
    const OpStaticInfo& si = analyses_[t.job].info[op_idx];
    if (!src.source.empty()) {
      // user source (ops/source.h): reads exact rows, CPU-resident
      Profiler::Scope s(inst.profiler, "load:source:" + src.source);
      SourceConfig cfg;
      cfg.args = src.args;
      cfg.profiler = inst.profiler;
      auto src_inst = source_registry().get(src.source).make(cfg);
      ElementVector elems;
      src_inst->read(rows, elems);
      SCA_CHECK(elems.size() == rows.size(),
                "source '" + src.source + "' produced " +
                    std::to_string(elems.size()) + " rows, wanted " +
                    std::to_string(rows.size()));
      for (auto& e : elems) outs[op_idx][si.output_columns[0]][e.index] = e;
      continue;
    }
    TableMetadata table = db_->get_table(src.table);
    const std::string& col_name = si.output_columns[0];
    Profiler::Scope s(inst.profiler, "load:" + src.table);

    if (si.output_types[0] == ColumnType::Video) {
      auto items = items_for_rows(table, rows);
      std::unordered_map<i32, VideoMetadata> vm;
      for (auto& ir : items)
        vm[ir.item] = read_video_metadata(*db_, table, src.column, ir.item);

      bool any_svc = false;
      for (auto& kv2 : vm) {
        // Decoder-slot dispatch: h264 tables are fully indexed
        // (ingest.cpp) but decoding needs the rocDecode/VCN hardware
        // decoder, which this image does not ship — fail loudly instead
        // of silently mis-reading (the reference shipped its NVDEC path
        // disabled the same way, evaluate_worker.cpp:90-93).
        SCA_CHECK(kv2.second.codec != "h264",
                  "table '" + src.table +
                      "' is H.264: decoding requires the rocDecode/VCN "
                      "hardware decoder (not in this image). Export it "
                      "with save_mp4() or transcode to codec='svc'. "
                      "Decoder slot: csrc/video/ingest.h");
        any_svc |= kv2.second.codec == "svc";
      }
      if (any_svc) {
        // Codec path (decoder-automaton parity): set up per-item decode
        // states; frames decode per work packet in advance_svc_source, on
        // the instance's GPU when it has one (straight into HBM).
        auto& states = svc_states[op_idx];
        size_t ri = 0;
        for (auto& ir : items) {
          const VideoMetadata& m = vm[ir.item];
          SvcItemState st;
          st.item = ir.item;
          st.row_start = ir.row_start;
          st.vm = m;
          while (ri < rows.size() && rows[ri] < ir.row_end) {
            st.local.push_back(rows[ri] - ir.row_start);
            ++ri;
          }
          std::vector<i64> span = svc_decode_span(m, st.local);
          if (span.empty()) continue;
          st.lo = m.sample_offsets[span.front()];
          st.hi = m.sample_offsets[span.back()] + m.sample_sizes[span.back()];
          // Device-resident bytes first (prefetched by the load worker or
          // resident from an earlier task), then prefetched host bytes,
          // then cache-fill / storage read.
          st.cached =
              prep ? prep->take_cached(op_idx, ir.item, st.lo, st.hi)
                   : nullptr;
          if (st.cached &&
              (!inst.gpu.is_gpu() || st.cached->dev.id != inst.gpu.id)) {
            st.cached.reset();  // prefetched for a different GPU
          }
          st.host =
              prep ? prep->take_span(op_idx, ir.item, st.lo, st.hi) : nullptr;
          if (!st.cached && inst.gpu.is_gpu()) {
            st.cached = acquire_ready_span(table, src.column, ir.item, st.lo,
                                           st.hi, inst.gpu, inst.profiler,
                                           st.host);
          }
          if (st.cached && st.host) {
            delete_buffer(CPU_DEVICE, st.host);
            st.host = nullptr;
          }
          if (!st.cached && !st.host) {
            st.host = new_buffer(CPU_DEVICE, st.hi - st.lo);
            Profiler::Scope sl(inst.profiler, "load:video");
            db_->storage()->read_range(
                db_->paths().item(table.id, table.column_id(src.column),
                                  ir.item),
                st.lo, st.hi - st.lo, st.host);
            inst.profiler->increment("io_read_bytes", (i64)(st.hi - st.lo));
          }
          states.push_back(std::move(st));
        }
        continue;
      }

      ElementVector elems =
          read_column_rows(*db_, table, src.column, rows,
                           pp_.sparsity_threshold);
      size_t ei = 0;
      for (auto& ir : items) {
        const VideoMetadata& m = vm[ir.item];
        SCA_CHECK(m.codec == "raw", "unknown video codec '" + m.codec + "'");
        while (ei < elems.size() && elems[ei].index < ir.row_end) {
          Element& e = elems[ei];
          e.is_frame = true;
          e.frame_info.shape[0] = m.height;
          e.frame_info.shape[1] = m.width;
          e.frame_info.shape[2] = m.channels;
          e.frame_info.type = m.frame_type;
          outs[op_idx][col_name][e.index] = e;
          ++ei;
        }
      }
    } else {
      ElementVector elems = read_column_rows(*db_, table, src.column, rows,
                                             pp_.sparsity_threshold);
      for (auto& e : elems) outs[op_idx][col_name][e.index] = e;
    }
  }
}

void LocalExecutor::advance_svc_source(Instance& inst,
                                       std::vector<SvcItemState>& states,
                                       i64 max_row, const std::string& col,
                                       OutsVec& outs, i32 op_idx) {
  for (auto& st : states) {
    size_t k = st.cursor;
    while (k < st.local.size() && st.local[k] + st.row_start <= max_row) ++k;
    if (k == st.cursor) continue;
    std::vector<i64> want(st.local.begin() + st.cursor,
                          st.local.begin() + k);
    st.cursor = k;
    if (st.cached) {
      Profiler::Scope sd(inst.profiler, "decode:gpu");
      inst.profiler->increment("decoded_frames", (i64)want.size());
      auto elems = svc_decode_gpu_dev(st.cached->ptr, st.lo, st.vm, want,
                                      inst.gpu);
      for (auto& e : elems) {
        e.index += st.row_start;
        outs[op_idx][col][e.index] = e;
      }
    } else if (inst.gpu.is_gpu()) {
      Profiler::Scope sd(inst.profiler, "decode:gpu");
      inst.profiler->increment("decoded_frames", (i64)want.size());
      auto elems =
          svc_decode_gpu(st.host, st.hi - st.lo, st.vm, want, inst.gpu,
                         st.lo);
      for (auto& e : elems) {
        e.index += st.row_start;
        outs[op_idx][col][e.index] = e;
      }
    } else {
      Profiler::Scope sd(inst.profiler, "decode:cpu");
      std::vector<std::vector<u8>> frames;
      svc_decode_cpu(st.host, st.hi - st.lo, st.vm, want, frames, st.lo);
      for (size_t q = 0; q < want.size(); ++q) {
        Element e;
        e.is_frame = true;
        e.frame_info.shape[0] = st.vm.height;
        e.frame_info.shape[1] = st.vm.width;
        e.frame_info.shape[2] = st.vm.channels;
        e.frame_info.type = st.vm.frame_type;
        e.size = frames[q].size();
        e.buffer = new_buffer(CPU_DEVICE, e.size);
        std::memcpy(e.buffer, frames[q].data(), e.size);
        e.index = want[q] + st.row_start;
        outs[op_idx][col][e.index] = e;
      }
    }
    // release the encoded bytes once the item is fully decoded (decode
    // synced its stream before returning)
    if (st.cursor == st.local.size()) {
      st.cached.reset();
      if (st.host) {
        delete_buffer(CPU_DEVICE, st.host);
        st.host = nullptr;
      }
    }
  }
}


void LocalExecutor::process_task(Instance& inst, const TaskDesc& t,
                                 PreparedTask* prep) {
  // HIP device selection is per-thread: instance threads (std::thread in
  // run(), Python threads in the distributed worker) start on device 0,
  // which is wrong for every rank but rank 0 on a multi-GPU node. Pin this
  // thread to the instance's GPU before any stream/kernel use (sticky on
  // purpose — the thread serves this instance for its lifetime).
  if (inst.gpu.is_gpu()) {
    (void)hipSetDevice(inst.gpu.id);
  }
  const JobBinding& job = jobs_[t.job];
  const JobAnalysis& ja = analyses_[t.job];
  size_t n = graph_.ops.size();

  TaskPlan plan;
  if (prep) {
    plan = std::move(prep->plan);
  } else {
    Profiler::Scope s(inst.profiler, "derive_task_plan");
    plan = derive_task_plan(graph_, ja, job, t.start, t.end);
  }

  // (Re)build kernel instances for this job's per-stream args.
  if (inst.kernels_job != t.job) {
    inst.kernels.clear();
    inst.kernels.resize(n);
    for (size_t i = 0; i < n; ++i) {
      const OpNode& op = graph_.ops[i];
      if (is_builtin_op(op.name)) continue;
      KernelConfig cfg;
      cfg.device = op.device == DeviceType::GPU ? inst.gpu : CPU_DEVICE;
      cfg.output_device = cfg.device;
      SCA_CHECK(op.device != DeviceType::GPU || cfg.device.is_gpu(),
                "graph requests GPU but executor has no gpu_ids");
      for (auto& e : op.inputs) cfg.input_columns.push_back(e.column);
      cfg.output_columns = ja.info[i].output_columns;
      cfg.args = op.args;
      cfg.node_id = (i32)i;
      cfg.max_batch = ja.info[i].batch;
      cfg.profiler = inst.profiler;
      const KernelFactory& kf = kernel_registry().get(op.name, op.device);
      if (kf.output_device_type >= 0) {
        cfg.output_device =
            kf.output_device_type == (i32)DeviceType::GPU
                ? inst.gpu
                : CPU_DEVICE;
      }
      inst.kernels[i] = kf.make(cfg);
      auto ait = job.op_args.find((i32)i);
      std::vector<u8> sargs =
          ait == job.op_args.end() ? std::vector<u8>{} : ait->second;
      // fetch-once resources (weights etc.) — first fetch per
      // (op name, args) within THIS executor (so a later job using the same
      // op with different args re-fetches; reference fetches per job,
      // evaluate_worker.cpp:493-550 worker-0 fetch + setup barrier).
      {
        std::string fetch_key(op.name);
        fetch_key.append(1, '\0');
        fetch_key.append(op.args.begin(), op.args.end());
        std::lock_guard<std::mutex> fl(fetch_mu_);
        if (!fetched_.count(fetch_key)) {
          inst.kernels[i]->fetch_resources(op.args);
          fetched_.insert(fetch_key);
        }
      }
      inst.kernels[i]->setup_with_resources(op.args);
      inst.kernels[i]->new_stream(sargs);
    }
    inst.kernels_job = t.job;
  }

  // ---- streaming packet execution with liveness-driven frees ----
  //
  // The task no longer materializes every op's whole-task output: sink rows
  // are processed in work_packet_size packets; a backward pass per packet
  // computes how far each op's cursor must advance (stencils/warmup/remaps
  // included — the per-task plan already resolved exact row sets), sources
  // decode incrementally, and every element is freed as soon as its last
  // planned consumer has read it. Peak engine memory is therefore bounded
  // by the live stencil window, independent of io_packet_size (reference
  // analogue: work packets flowing through stage threads,
  // worker.cpp:1663-1722, + liveness dag_analysis.cpp:1145-1326).
  OutsVec outs(n);
  // planned read counts per (op, column, row)
  std::vector<std::map<std::string, std::unordered_map<i64, i32>>> reads(n);
  const OpNode& out_op = graph_.ops.back();
  for (size_t j = 0; j < n; ++j) {
    const OpNode& op = graph_.ops[j];
    const OpTaskPlan& otp = plan.ops[j];
    if (is_input_op(op.name)) continue;
    if (is_output_op(op.name)) {
      for (auto& e : op.inputs) {
        auto& rm = reads[e.op][e.column];
        for (i64 row = t.start; row < t.end; ++row) rm[row]++;
      }
    } else if (is_sample_op(op.name) || is_slice_op(op.name) ||
               is_unslice_op(op.name)) {
      if (op.inputs.empty()) continue;
      auto& rm = reads[op.inputs[0].op][op.inputs[0].column];
      for (i64 u : otp.remap)
        if (u >= 0) rm[u]++;
    } else {
      for (auto& e : op.inputs) {
        auto& rm = reads[e.op][e.column];
        for (auto& win : otp.windows)
          for (i64 w : win) rm[w]++;
      }
    }
  }

  std::vector<std::pair<DeviceHandle, u8*>> pending_free;
  auto dec_read = [&](i32 opi, const std::string& col, i64 row) {
    auto& rm = reads[opi][col];
    auto it = rm.find(row);
    SCA_CHECK(it != rm.end(), "liveness underflow (op " +
                                  graph_.ops[opi].name + " row " +
                                  std::to_string(row) + ")");
    if (--it->second == 0) {
      rm.erase(it);
      auto& om = outs[opi][col];
      auto oit = om.find(row);
      if (oit != om.end()) {
        if (oit->second.buffer)
          pending_free.emplace_back(oit->second.device, oit->second.buffer);
        om.erase(oit);
      }
    }
  };
  // Frees are deferred to packet boundaries: same-stream ordering covers
  // this instance's later work, but the pool is shared across instances,
  // so wait for in-flight kernels before returning buffers.
  auto flush_frees = [&]() {
    if (pending_free.empty()) return;
    if (inst.gpu.is_gpu()) sync_per_thread_stream();
    for (auto& pf : pending_free) delete_buffer(pf.first, pf.second);
    pending_free.clear();
  };

  // Sink accumulation (one storage item per task, written at task end;
  // D2H happens per packet so it overlaps later compute).
  std::vector<std::vector<Element>> sink_cols(out_op.inputs.size());
  std::vector<std::vector<u8>> sink_owned(out_op.inputs.size());

  std::map<i32, std::vector<SvcItemState>> svc_states;
  std::vector<size_t> cursor(n, 0);

  auto cleanup_all = [&]() {
    if (inst.gpu.is_gpu()) {
      try {
        sync_per_thread_stream();
      } catch (...) {
      }
    }
    for (auto& pf : pending_free) delete_buffer(pf.first, pf.second);
    pending_free.clear();
    for (auto& per_op : outs) {
      for (auto& per_col : per_op) {
        for (auto& kv : per_col.second) {
          Element& e = kv.second;
          if (e.buffer) delete_buffer(e.device, e.buffer);
        }
        per_col.second.clear();
      }
    }
    for (size_t c = 0; c < sink_cols.size(); ++c) {
      for (size_t k = 0; k < sink_cols[c].size(); ++k) {
        if (sink_owned[c][k] && sink_cols[c][k].buffer)
          delete_buffer(CPU_DEVICE, sink_cols[c][k].buffer);
      }
      sink_cols[c].clear();
      sink_owned[c].clear();
    }
    for (auto& kv : svc_states) {
      for (auto& st : kv.second) {
        st.cached.reset();
        if (st.host) {
          delete_buffer(CPU_DEVICE, st.host);
          st.host = nullptr;
        }
      }
    }
  };

  try {
    init_sources(inst, t, plan, outs, svc_states, prep);

    i64 W = std::max<i64>(1, pp_.work_packet_size);
    for (i64 ps = t.start; ps < t.end; ps += W) {
      i64 pe = std::min(t.end, ps + W);

      // backward pass: per-op cursor targets for this packet
      std::vector<i64> need(n, -1);
      std::vector<size_t> target(n, 0);
      for (auto& e : out_op.inputs)
        need[e.op] = std::max(need[e.op], pe - 1);
      for (i64 i = (i64)n - 2; i >= 0; --i) {
        const OpNode& op = graph_.ops[i];
        const OpTaskPlan& otp = plan.ops[i];
        if (is_input_op(op.name)) continue;
        if (is_sample_op(op.name) || is_slice_op(op.name) ||
            is_unslice_op(op.name)) {
          const auto& rr = otp.required_rows;
          size_t tgt =
              std::upper_bound(rr.begin(), rr.end(), need[i]) - rr.begin();
          if (tgt < cursor[i]) tgt = cursor[i];
          for (size_t k = cursor[i]; k < tgt; ++k) {
            if (otp.remap[k] >= 0)
              need[op.inputs[0].op] =
                  std::max(need[op.inputs[0].op], otp.remap[k]);
          }
          target[i] = tgt;
        } else {
          const auto& cr = otp.compute_rows;
          size_t tgt =
              std::upper_bound(cr.begin(), cr.end(), need[i]) - cr.begin();
          if (tgt < cursor[i]) tgt = cursor[i];
          if (tgt > cursor[i]) {
            i64 mx = -1;
            for (size_t k = cursor[i]; k < tgt; ++k)
              for (i64 w : otp.windows[k]) mx = std::max(mx, w);
            if (mx >= 0)
              for (auto& e : op.inputs)
                need[e.op] = std::max(need[e.op], mx);
          }
          target[i] = tgt;
        }
      }

      // forward pass
      for (size_t i = 0; i < n; ++i) {
        const OpNode& op = graph_.ops[i];
        const OpStaticInfo& si = ja.info[i];
        OpTaskPlan& otp = plan.ops[i];

        if (is_input_op(op.name)) {
          if (need[i] >= 0) {
            auto sit = svc_states.find((i32)i);
            if (sit != svc_states.end())
              advance_svc_source(inst, sit->second, need[i],
                                 si.output_columns[0], outs, (i32)i);
            // eager sources already materialized in init_sources
          }
          continue;
        }

        if (is_sample_op(op.name) || is_slice_op(op.name) ||
            is_unslice_op(op.name)) {
          // remap: alias parent elements (zero-copy, extra ref per alias)
          auto& parent_map = outs[op.inputs[0].op][op.inputs[0].column];
          auto& out_map = outs[i][si.output_columns[0]];
          for (size_t k = cursor[i]; k < target[i]; ++k) {
            i64 row = otp.required_rows[k];
            i64 up = otp.remap[k];
            Element e;
            if (up < 0) {
              e.is_null = true;
            } else {
              auto pit = parent_map.find(up);
              SCA_CHECK(pit != parent_map.end(),
                        "remap source row missing (op " + op.name + ")");
              e = pit->second;
              if (e.buffer) add_buffer_ref(e.device, e.buffer);
            }
            e.index = row;
            out_map[row] = e;
            if (up >= 0) dec_read(op.inputs[0].op, op.inputs[0].column, up);
          }
          cursor[i] = target[i];
          continue;
        }

        if (is_output_op(op.name)) {
          // gather packet rows, batched D2H, hold refs for the task-end
          // write
          Profiler::Scope s(inst.profiler, "save");
          std::vector<u8*> d2h_dst;
          std::vector<const u8*> d2h_src;
          std::vector<size_t> d2h_sz;
          DeviceHandle gpu_src = CPU_DEVICE;
          for (size_t c = 0; c < op.inputs.size(); ++c) {
            auto& parent_map = outs[op.inputs[c].op][op.inputs[c].column];
            for (i64 row = ps; row < pe; ++row) {
              auto pit = parent_map.find(row);
              SCA_CHECK(pit != parent_map.end(), "sink missing row");
              Element e = pit->second;
              if (e.is_null) {
                Element ne;
                ne.is_null = true;
                ne.index = row;
                sink_cols[c].push_back(ne);
                sink_owned[c].push_back(0);
                continue;
              }
              if (!e.device.is_gpu()) {
                // producer element may be freed before the task-end write;
                // the sink keeps its own block reference
                add_buffer_ref(e.device, e.buffer);
                e.index = row;
                sink_cols[c].push_back(e);
                sink_owned[c].push_back(1);
                continue;
              }
              Element ce = e;
              ce.device = CPU_DEVICE;
              ce.buffer = new_buffer(CPU_DEVICE, e.size);
              ce.index = row;
              sink_cols[c].push_back(ce);
              sink_owned[c].push_back(1);
              d2h_dst.push_back(ce.buffer);
              d2h_src.push_back(e.buffer);
              d2h_sz.push_back(e.size);
              gpu_src = e.device;
            }
          }
          if (!d2h_dst.empty()) {
            memcpy_vec(d2h_dst, CPU_DEVICE, d2h_src, gpu_src, d2h_sz);
          }
          for (size_t c = 0; c < op.inputs.size(); ++c)
            for (i64 row = ps; row < pe; ++row)
              dec_read(op.inputs[c].op, op.inputs[c].column, row);
          continue;
        }

        // ---- kernel op: advance [cursor, target) in batches ----
        BaseKernel* kernel = inst.kernels[i].get();
        DeviceHandle kdev = kernel->config().device;
        auto& out_maps = outs[i];
        auto rit = reads[i].end();  // re-looked-up per column below

        size_t bi = cursor[i];
        while (bi < target[i]) {
          size_t be = bi + 1;
          while (be < target[i] && (i64)(be - bi) < (i64)si.batch &&
                 !otp.reset_before[be]) {
            ++be;
          }
          if (otp.reset_before[bi]) kernel->reset();

          StenciledElements input(op.inputs.size());
          std::vector<Element> scratch;  // owned device copies to free
          for (size_t c = 0; c < op.inputs.size(); ++c) {
            auto& parent_map = outs[op.inputs[c].op][op.inputs[c].column];
            input[c].resize(be - bi);
            for (size_t r = bi; r < be; ++r) {
              for (i64 wrow : otp.windows[r]) {
                auto pit = parent_map.find(wrow);
                SCA_CHECK(pit != parent_map.end(),
                          "kernel input row missing for op '" + op.name +
                              "'");
                bool copied;
                Element e = element_to_device(pit->second, kdev, copied);
                if (copied) scratch.push_back(e);
                input[c][r - bi].push_back(e);
              }
            }
          }

          BatchedElements output(si.output_columns.size());
          {
            Profiler::Scope s(inst.profiler, "op:" + op.name);
            kernel->execute(input, output);
          }
          for (size_t c = 0; c < si.output_columns.size(); ++c) {
            SCA_CHECK(output[c].size() == be - bi,
                      "op '" + op.name + "' produced " +
                          std::to_string(output[c].size()) +
                          " rows, expected " + std::to_string(be - bi));
            rit = reads[i].find(si.output_columns[c]);
            auto& out_map = out_maps[si.output_columns[c]];
            for (size_t r = bi; r < be; ++r) {
              Element& e = output[c][r - bi];
              e.index = otp.compute_rows[r];
              if (!e.is_null) e.device = kernel->config().output_device;
              bool wanted =
                  rit != reads[i].end() && rit->second.count(e.index);
              if (wanted) {
                out_map[e.index] = e;
              } else if (e.buffer) {
                // warmup-only row or unused output column — freed at the
                // packet boundary (after the stream sync)
                pending_free.emplace_back(e.device, e.buffer);
              }
            }
          }
          // GPU kernels are async on this thread's stream; staging copies
          // can only be released after the kernels that read them complete.
          if (!scratch.empty() && kdev.is_gpu()) sync_per_thread_stream();
          for (auto& e : scratch) delete_buffer(e.device, e.buffer);
          // consume the stencil reads of this batch
          for (size_t c = 0; c < op.inputs.size(); ++c)
            for (size_t r = bi; r < be; ++r)
              for (i64 wrow : otp.windows[r])
                dec_read(op.inputs[c].op, op.inputs[c].column, wrow);
          bi = be;
        }
        cursor[i] = target[i];
      }

      flush_frees();
    }

    // ---- task-end write: one storage item per (column, task) ----
    if (!job.sink.sink.empty()) {
      // user sink (ops/source.h): bypasses table storage entirely
      Profiler::Scope s(inst.profiler, "save:sink:" + job.sink.sink);
      SinkConfig cfg;
      cfg.args = job.sink.args;
      cfg.profiler = inst.profiler;
      auto sink_inst = sink_registry().get(job.sink.sink).make(cfg);
      sink_inst->new_task(t.task);
      sink_inst->write(sink_cols);
      sink_inst->finished();
      for (size_t c = 0; c < sink_cols.size(); ++c) {
        for (size_t k = 0; k < sink_cols[c].size(); ++k) {
          if (sink_owned[c][k] && sink_cols[c][k].buffer)
            delete_buffer(CPU_DEVICE, sink_cols[c][k].buffer);
        }
        sink_cols[c].clear();
        sink_owned[c].clear();
      }
    } else {
      Profiler::Scope s(inst.profiler, "save");
      const TableMetadata& table = out_tables_[t.job];
      auto& cols = sink_cols;
      auto& owned = sink_owned;
      // Sink-side codec annotations (OpColumn.compress_video; parity:
      // reference compressed-output columns + PostEvaluateWorker encode,
      // evaluate_worker.cpp:1329-1560).
      std::map<std::string, std::string> compress;
      if (!out_op.args.empty()) {
        auto a = mp::decode(out_op.args);
        auto& am = a.as_map();
        auto cit = am.find("compress");
        if (cit != am.end()) {
          for (auto& kv : cit->second.as_map())
            compress[kv.first] = kv.second.as_str();
        }
      }
      auto write_col = [&](size_t c) {
        const std::string& cname = table.columns[c].name;
        auto cmp = compress.find(cname);
        if (cmp != compress.end() &&
            table.columns[c].type == ColumnType::Video) {
          SCA_CHECK(cmp->second == "svc",
                    "unknown sink codec '" + cmp->second + "'");
          i32 h = 0, w = 0, ch = 0;
          size_t fsize = 0;
          std::vector<u8> contig;
          i64 nf = 0;
          for (auto& e : cols[c]) {
            SCA_CHECK(!e.is_null,
                      "cannot codec-compress a column with null rows");
            SCA_CHECK(e.is_frame && e.frame_info.type == FrameType::U8,
                      "svc compression needs u8 frames");
            if (nf == 0) {
              h = e.frame_info.shape[0];
              w = e.frame_info.shape[1];
              ch = e.frame_info.shape[2];
              fsize = e.size;
              contig.reserve(fsize * cols[c].size());
            }
            SCA_CHECK(e.size == fsize,
                      "svc compression needs uniform frame sizes");
            contig.insert(contig.end(), e.buffer, e.buffer + e.size);
            ++nf;
          }
          VideoMetadata vm;
          vm.width = w;
          vm.height = h;
          vm.channels = ch;
          vm.frame_type = FrameType::U8;
          vm.num_frames = nf;
          std::vector<u8> stream;
          svc_encode_cpu(contig.data(), nf, h, w, ch, /*gop=*/16, stream,
                         vm);
          inst.profiler->increment("io_write_bytes", (i64)stream.size());
          write_video_item(*db_, table, cname, t.task, stream, vm);
          for (size_t k = 0; k < cols[c].size(); ++k) {
            if (owned[c][k] && cols[c][k].buffer)
              delete_buffer(CPU_DEVICE, cols[c][k].buffer);
          }
          cols[c].clear();
          owned[c].clear();
          return;
        }
        {
          i64 wb = 0;
          for (auto& e : cols[c]) wb += (i64)e.size;
          inst.profiler->increment("io_write_bytes", wb);
        }
        write_column_item(*db_, table, table.columns[c].name, t.task,
                          cols[c]);
        if (table.columns[c].type == ColumnType::Video) {
          // Raw-stored frame column: record geometry so readers can
          // reconstruct frames.
          VideoMetadata vm;
          vm.codec = "raw";
          vm.num_frames = (i64)cols[c].size();
          for (auto& e : cols[c]) {
            if (e.is_null) continue;
            vm.height = e.frame_info.shape[0];
            vm.width = e.frame_info.shape[1];
            vm.channels = e.frame_info.shape[2];
            vm.frame_type = e.frame_info.type;
            break;
          }
          for (i64 k = 0; k < vm.num_frames; ++k) {
            vm.keyframe_indices.push_back(k);
            vm.sample_sizes.push_back(cols[c][k].size);
            vm.sample_offsets.push_back(
                k == 0 ? 0 : vm.sample_offsets[k - 1] +
                                 vm.sample_sizes[k - 1]);
          }
          auto vbuf = vm.serialize();
          db_->storage()->write_all(
              db_->paths().video_metadata(
                  table.id, table.column_id(table.columns[c].name), t.task),
              vbuf.data(), vbuf.size());
        }
        for (size_t k = 0; k < cols[c].size(); ++k) {
          if (owned[c][k] && cols[c][k].buffer)
            delete_buffer(CPU_DEVICE, cols[c][k].buffer);
        }
        cols[c].clear();
        owned[c].clear();
      };
      // Save fan-out (reference: SaveWorker multi-sink thread pool,
      // save_worker.cpp:104-140): independent output columns write and
      // codec-compress in parallel.
      if (out_op.inputs.size() <= 1) {
        if (!out_op.inputs.empty()) write_col(0);
      } else {
        std::vector<std::thread> savers;
        std::exception_ptr save_err;
        std::mutex err_mu;
        for (size_t c = 0; c < out_op.inputs.size(); ++c) {
          savers.emplace_back([&, c] {
            try {
              write_col(c);
            } catch (...) {
              std::lock_guard<std::mutex> el(err_mu);
              if (!save_err) save_err = std::current_exception();
            }
          });
        }
        for (auto& th : savers) th.join();
        if (save_err) std::rethrow_exception(save_err);
      }
    }
  } catch (...) {
    cleanup_all();
    throw;
  }
  cleanup_all();
  inst.profiler->increment("tasks", 1);
  inst.profiler->increment("rows", t.end - t.start);
  tasks_done_.fetch_add(1);
}

}  // namespace sca
