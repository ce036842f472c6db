// Single-node executor: turns (graph, job bindings) into committed output
// tables. Capability parity: the worker-side execution path of the
// reference (worker.cpp process_job + load/pre-evaluate/evaluate/
// post-evaluate/save workers). MI355X layout: one pipeline instance per GPU
// (or N CPU instances), each instance a worker thread owning a HIP stream;
// kernels and device copies are issued async on that stream and only the
// sink synchronizes. Tasks are pulled from a shared queue (the in-process
// analogue of the master's pull scheduler; the distributed version reuses
// process_task via the worker service).
#pragma once

#include <atomic>
#include <mutex>
#include <set>
#include <thread>

#include "../dag/graph.h"
#include "../ops/kernel.h"
#include "../profiler.h"
#include "../queue.h"
#include "../video/span_cache.h"
#include "table_io.h"

namespace sca {

struct PerfParams {
  i64 io_packet_size = 128;
  i64 work_packet_size = 32;
  i32 pipeline_instances = 1;
  i32 load_workers = 0;  // 0 = auto (min(8, max(2, instances)))
  i64 queue_size = 4;
  size_t cpu_pool_size = 0;
  size_t gpu_pool_size = 0;
  i32 sparsity_threshold = 8;
  i32 profiler_level = 1;
  // HBM span-cache byte budget (compressed input GOP spans stay resident in
  // device memory; video/span_cache.h). 0 = auto: gpu_pool/3 capped at
  // 16 GB when a GPU pool exists, else 4 GB. Set to 1 to disable.
  size_t span_cache_size = 0;
};

struct TaskDesc {
  i32 job = 0;
  i32 task = 0;  // item index in the output table
  i64 start = 0, end = 0;
};

// A task with its plan derived and its video source byte spans already
// read into pinned memory by a load worker (reference: the dedicated
// load_worker threads feeding the pipeline, worker.cpp:85 +
// load_worker.cpp). Owns the span buffers until consumed.
struct PreparedTask {
  TaskDesc desc;
  TaskPlan plan;
  struct Span {
    i32 op = 0;
    i32 item = 0;
    u64 lo = 0, hi = 0;
    u8* buf = nullptr;     // pinned (CPU pool); null when cached on-device
    SpanHandle cached;     // HBM-resident encoded bytes (span cache)
  };
  std::vector<Span> spans;
  ~PreparedTask();
  u8* take_span(i32 op, i32 item, u64 lo, u64 hi);
  SpanHandle take_cached(i32 op, i32 item, u64 lo, u64 hi);
};

class LocalExecutor {
 public:
  LocalExecutor(std::shared_ptr<Database> db, JobGraph graph,
                std::vector<JobBinding> jobs, PerfParams pp,
                std::vector<i32> gpu_ids);
  ~LocalExecutor();

  // Analyze + create output tables + run all tasks to completion. Throws on
  // the first failure.
  void run();

  // Process one externally-scheduled task (distributed mode). Analysis and
  // output tables must exist (call prepare() first). Workers pass
  // create_outputs=false: the master already created the output tables and
  // they attach to the existing descriptors (shared storage).
  void prepare(bool create_outputs = true);
  void process_task_public(i32 instance, const TaskDesc& t);
  // Distributed-worker prefetch: load-worker half on any thread, then the
  // instance half (same split run() uses internally).
  std::shared_ptr<PreparedTask> prepare_task_public(const TaskDesc& t);
  void process_prepared_public(i32 instance,
                               const std::shared_ptr<PreparedTask>& pt);
  std::vector<TaskDesc> all_tasks() const;
  void finalize_job(i32 job);  // set end_rows + commit output table

  const std::vector<std::unique_ptr<Profiler>>& profilers() const {
    return profilers_;
  }
  i64 total_output_rows() const;
  i64 tasks_done() const { return tasks_done_.load(); }

 private:
  struct Instance;
  using OutsVec =
      std::vector<std::map<std::string, std::unordered_map<i64, Element>>>;
  // Incremental decode state for one svc video item of one Input op: the
  // encoded byte span (device-cached or pinned host) plus a cursor over the
  // wanted item-local rows, so decode advances one work packet at a time
  // as the streaming executor pulls rows (reference analogue: the
  // feeder/retriever DecoderAutomata, decoder_automata.cpp:126-230).
  struct SvcItemState {
    i32 item = 0;
    i64 row_start = 0;
    VideoMetadata vm;
    std::vector<i64> local;  // item-local wanted rows, ascending
    size_t cursor = 0;       // decoded prefix of `local`
    SpanHandle cached;       // HBM-resident encoded bytes [lo, hi)
    u8* host = nullptr;      // pinned encoded bytes [lo, hi)
    u64 lo = 0, hi = 0;
  };
  void process_task(Instance& inst, const TaskDesc& t,
                    PreparedTask* prep = nullptr);
  // Load-worker half of a task: derive the plan + read video source spans
  // into pinned buffers (runs on prefetch threads in run()).
  std::shared_ptr<PreparedTask> prepare_task(const TaskDesc& t,
                                             Profiler* prof);
  // Eagerly load non-video / raw-codec source rows into `outs`; set up
  // incremental decode states for svc video sources.
  void init_sources(Instance& inst, const TaskDesc& t, const TaskPlan& plan,
                    OutsVec& outs,
                    std::map<i32, std::vector<SvcItemState>>& svc_states,
                    PreparedTask* prep);
  // Decode all not-yet-decoded rows with global index <= max_row.
  void advance_svc_source(Instance& inst, std::vector<SvcItemState>& states,
                          i64 max_row, const std::string& col, OutsVec& outs,
                          i32 op_idx);
  void make_instance(i32 idx);
  // Lookup-or-fill the HBM span cache for one encoded byte range; returns a
  // ready handle (device-resident bytes) or null when caching declined.
  // `host_bytes`, when non-null, provides the bytes (skips the storage
  // read on a fresh insert).
  SpanHandle acquire_ready_span(const TableMetadata& table,
                                const std::string& column, i32 item, u64 lo,
                                u64 hi, DeviceHandle gpu, Profiler* prof,
                                const u8* host_bytes);

  std::shared_ptr<Database> db_;
  JobGraph graph_;
  std::vector<JobBinding> jobs_;
  PerfParams pp_;
  std::vector<i32> gpu_ids_;

  u64 db_key_ = 0;  // hash of db root, part of span-cache keys
  std::vector<JobAnalysis> analyses_;
  std::vector<TableMetadata> out_tables_;   // per job
  std::vector<std::vector<i64>> task_rows_; // per job: cumulative task ends
  bool prepared_ = false;

  std::atomic<i64> tasks_done_{0};
  std::vector<std::unique_ptr<Instance>> instances_;
  std::vector<std::unique_ptr<Profiler>> profilers_;

  // fetch_resources dedup, scoped to this executor: keyed by
  // op name + args so same-op-different-args jobs each fetch.
  std::mutex fetch_mu_;
  std::set<std::string> fetched_;
};

// Move an element to the target device; returns the original element if
// already resident (borrowed == true) else an owned copy.
Element element_to_device(const Element& e, DeviceHandle target, bool& copied);

}  // namespace sca
