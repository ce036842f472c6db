#!/usr/bin/env python3
"""Flagship benchmark: frames/sec (whole node) on the BASELINE.json
pipelines — 1080p histogram (SVC GPU decode + HIP histogram) and ResNet-50
frame classification (bf16 MFMA) — on synthetic video with random-init
weights (no network in this environment; BASELINE.md records that the
reference repo publishes no in-repo numbers, so these runs establish the
recorded baseline).

Launched by the driver as:
  python bench.py --gpus N --steps K --warmup W          (N=1, direct)
  torchrun --nproc-per-node N bench.py --gpus N ...      (N>1, one rank/GPU)

One step = processing FRAMES_PER_STEP 1080p frames per GPU through the
engine pipeline (load -> SVC GPU decode -> op(s) -> save). Weak scaling:
per-GPU work is fixed as N grows. Rank 0 prints one JSON line.
"""
import argparse
import json
import os
import shutil
import sys
import tempfile
import time

import numpy as np

FRAMES_PER_STEP = 512
H, W = 1080, 1920


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def make_clip(n, h=H, w=W):
    """Synthetic 1080p clip: smooth moving gradient + mild texture, the
    shape/codec-behavior of real footage (temporally coherent)."""
    yy, xx = np.mgrid[0:h, 0:w]
    rng = np.random.RandomState(0)
    texture = rng.randint(0, 32, size=(h, w, 3)).astype(np.int32)
    frames = np.zeros((n, h, w, 3), np.uint8)
    for i in range(n):
        frames[i, :, :, 0] = (xx + i * 2 + texture[:, :, 0]) % 256
        frames[i, :, :, 1] = (yy + i + texture[:, :, 1]) % 256
        frames[i, :, :, 2] = (xx + yy + i * 3 + texture[:, :, 2]) % 256
    return frames


def build_pipeline(sc, sp, video, pipeline, device, out_name):
    frame = sc.io.Input([video])
    cols = []
    # batch=64 measured best for the flagship (18.1k vs 16.4k at 16):
    # bigger GEMM M amortizes per-launch cost across the 53-conv forward
    dnn_batch = int(os.environ.get("SCANNER_BENCH_BATCH", "64"))
    if pipeline in ("hist", "full"):
        cols.append(sc.ops.Histogram(frame=frame, device=device))
    if pipeline in ("resnet", "full"):
        cols.append(sc.ops.ResNet50(frame=frame, device=device,
                                    batch=dnn_batch))
    if pipeline == "flow":
        # BASELINE config 4: dense optical flow; per-frame flow summary is
        # the saved column (the 16 MB/frame flow field stays on-GPU).
        flow = sc.ops.OpticalFlow(frame=frame, device=device)
        cols.append(sc.ops.FlowStats(flow=flow, device=device))
    if pipeline == "pose":
        # BASELINE config 5: multi-DNN graph on 4K — pose keypoints and
        # ResNet-50 classification of every frame, decode+DNN overlapped.
        cols.append(sc.ops.Pose(frame=frame, device=device))
        cols.append(sc.ops.ResNet50(frame=frame, device=device))
    out = sp.NamedStream(sc, out_name)
    return sc.io.Output(cols, [out])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--pipeline", default="full",
                    choices=["hist", "resnet", "full", "flow", "pose"])
    ap.add_argument("--frames-per-step", type=int, default=None)
    args = ap.parse_args()

    import torch
    import torch.distributed as dist

    from scanner_amd import parallel

    rank, world, dist_device = parallel.init_from_env()
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1

    import scanner_amd as sp
    from scanner_amd import _core

    have_gpu = _core.have_gpu()
    device = sp.DeviceType.GPU if have_gpu else sp.DeviceType.CPU
    if args.pipeline in ("resnet", "full", "pose") and not (
            have_gpu and "ResNet50" in _core.registered_ops()):
        log("DNN ops need a GPU; falling back to hist pipeline")
        args.pipeline = "hist"
    if args.pipeline == "flow" and not _core.has_kernel(
            "OpticalFlow", int(device)):
        log("OpticalFlow kernel missing; falling back to hist")
        args.pipeline = "hist"

    tmp = tempfile.mkdtemp(prefix=f"scanner_bench_r{rank}_")
    sc = sp.Client(db_path=os.path.join(tmp, "db"))

    # pose runs on 4K per BASELINE config 5; everything else on 1080p
    h, w = (2160, 3840) if args.pipeline == "pose" else (H, W)
    n_frames = args.frames_per_step or \
        (128 if args.pipeline == "pose" else FRAMES_PER_STEP)
    log(f"[rank {rank}] ingesting {n_frames} synthetic {h}x{w} frames (svc)")
    clip = make_clip(n_frames, h=h, w=w)
    video = sp.NamedVideoStream(sc, "bench_clip", frames=clip,
                                codec="svc", io_packet_size=128)
    del clip

    # rank -> GPU modulo device count: the driver runs one rank per GPU
    # (identity mapping on an 8-GPU node); oversubscribed runs (2 ranks on
    # a 1-GPU box) share device 0 for RCCL-path shakeout
    gpu_ids = [local_rank % max(1, _core.gpu_device_count())] \
        if have_gpu else []
    # Pools: steady-state allocation must never hit the driver (hipMalloc
    # synchronizes the device; hipHostMalloc is ~ms per call).
    perf = sp.PerfParams.manual(
        work_packet_size=int(os.environ.get("SCANNER_BENCH_WORK", "32")),
        # io=128 over 64: fewer, larger tasks cut per-task scheduling and
        # span lookups (hist 55.8k vs 49.8k same box; flagship also +1%)
        io_packet_size=int(os.environ.get("SCANNER_BENCH_IO", "128")),
        gpu_pool=(32 << 30) if have_gpu else 0,
        cpu_pool=(4 << 30) if have_gpu else 0,
        # compressed input spans stay HBM-resident across steps (the whole
        # clip is ~2 GB encoded; 288 GB HBM) — storage+PCIe are paid once
        span_cache=int(os.environ.get("SCANNER_BENCH_SPANCACHE",
                                      str(8 << 30))) if have_gpu else 0)
    # measured sweet spots (r02 A/B): compute-saturated pipelines
    # (full/resnet/flow/pose) peak at 4 instances — more thrash the CUs
    # (inst=4 16.4k vs inst=8 14.7k f/s on the flagship); the decode-bound
    # hist pipeline still wants 6 to overlap decode chains.
    default_inst = "6" if args.pipeline == "hist" else "4"
    instances = int(os.environ.get("SCANNER_BENCH_INSTANCES",
                                   default_inst if have_gpu else "1"))

    def one_step(tag):
        sink = build_pipeline(sc, sp, video, args.pipeline, device,
                              f"bench_out_{tag}")
        sc.run(sink, perf, cache_mode=sp.CacheMode.Overwrite,
               gpu_ids=gpu_ids, pipeline_instances=instances)

    def sync():
        if have_gpu:
            torch.cuda.synchronize()
        if distributed:
            dist.barrier()

    for i in range(args.warmup):
        one_step(f"w{i}")
        log(f"[rank {rank}] warmup {i} done")

    sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(f"s{i}")
    sync()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if os.environ.get("SCANNER_PROFILE"):
        stats = sc.profile().statistics()
        for k in sorted(stats, key=lambda k: -stats[k]["total_ms"]):
            log(f"[prof] {k}: {stats[k]['total_ms']:.1f} ms "
                f"x{stats[k]['count']}")
    trace_path = os.environ.get("SCANNER_PROFILE_TRACE")
    if trace_path and rank == 0:
        # Chrome trace of the last step's pipeline stages (open in
        # chrome://tracing or Perfetto; parity: reference profiler
        # write_trace, profiler.py:57-198)
        sc.profile().write_trace(trace_path)
        log(f"[prof] chrome trace -> {trace_path}")
    elapsed = parallel.allreduce_max_time(elapsed, dist_device)

    total_frames = args.steps * n_frames * world
    fps = total_frames / elapsed
    if rank == 0:
        pipe_name = {"full": "histogram+ResNet50", "pose": "Pose+ResNet50",
                     }.get(args.pipeline, args.pipeline)
        res_name = "4K" if args.pipeline == "pose" else "1080p"
        # the flagship reports BASELINE.json's metric string verbatim
        metric = ("frames/sec (whole node), 1080p histogram + ResNet-50 "
                  "classify pipelines" if args.pipeline == "full" else
                  f"frames/sec (whole node), {res_name} "
                  f"{pipe_name} pipeline")
        result = {
            "metric": metric,
            "value": fps,
            "unit": "frames/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": {"resnet": "bf16", "full": "bf16", "pose": "bf16",
                      "flow": "f32", "hist": "u8"}[args.pipeline],
            "data": f"synthetic {res_name} video (svc-encoded), "
                    "random-init weights",
            "config": {
                "model": {"full": "1080p histogram + ResNet-50 classify",
                          "pose": "4K 3-stage 2-branch pose CNN + ResNet-50",
                          }.get(args.pipeline, args.pipeline),
                "global_batch": n_frames * world,
                "seq_len": n_frames,
                "parallelism": f"frame-shard dp{world}",
            },
        }
        print(json.dumps(result), flush=True)

    if distributed:
        # Post-timing RCCL gather over xGMI: each rank ships the first
        # result rows of its last step's output column to rank 0 (the
        # reference round-trips results through shared storage instead).
        # Runs AFTER the JSON line so a data-plane hiccup can never lose
        # the measurement; failures are logged, not fatal.
        try:
            sample = list(sp.NamedStream(
                sc, f"bench_out_s{args.steps-1}").load(rows=range(4)))
            gathered = parallel.gather_column(sample, dist_device)
            if rank == 0:
                log(f"gathered {len(gathered)} result rows over "
                    f"{dist.get_backend()}"
                    f"{' (RCCL/xGMI)' if dist.get_backend() == 'nccl' else ''}")
        except Exception as e:  # pragma: no cover
            log(f"[rank {rank}] post-bench gather failed: {e}")
        dist.destroy_process_group()
    shutil.rmtree(tmp, ignore_errors=True)


if __name__ == "__main__":
    main()
