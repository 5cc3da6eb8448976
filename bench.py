#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 bf16 synthetic ImageNet, images/sec
(BASELINE.json metric: "images/sec (whole node) ResNet-50 bf16 at
np=1/2/4/8").

Run single GPU:    python bench.py --gpus 1 --steps 20 --warmup 5
Run N GPUs (driver):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Weak scaling: per-GPU batch is fixed; `value` is the whole-job aggregate
images/sec over all ranks, using the MAX per-rank elapsed time.
"""

import argparse
import json
import os
import time

import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=512,
                    help="per-GPU batch size (weak scaling)")
    ap.add_argument("--model", default="resnet50",
                    choices=["resnet50", "bert", "gbt"])
    ap.add_argument("--no-graph", action="store_true",
                    help="disable hipGraph step capture")
    ap.add_argument("--allreduce-bench", action="store_true",
                    help="instead of a model step, sweep all-reduce "
                         "sizes and report bus bandwidth vs the "
                         "7x153 GB/s xGMI roofline")
    ap.add_argument("--bucket-mb", type=int, default=0,
                    help="override the DistributedOptimizer gradient "
                         "bucket size (sets SPARKDL_BUCKET_MB)")
    return ap.parse_args()


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, flush=True)


def _load_tunableop():
    """Load pre-captured hipBLASLt GEMM tunings for gfx950 (committed
    CSV; captured once with PYTORCH_TUNABLEOP_TUNING=1). Tuning itself
    stays off so fresh boxes pay no autotune cost."""
    if os.environ.get("PYTORCH_TUNABLEOP_ENABLED") is not None:
        return  # explicit env wins (capture runs)
    tune_file = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                             "tunableop_gfx950.csv")
    if not os.path.exists(tune_file):
        return
    try:
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(tune_file)
        log("TunableOp: loaded %s" % tune_file)
    except Exception as e:  # pragma: no cover
        log("TunableOp load failed: %s" % (e,))


def build_resnet_step(args, device, use_cuda):
    """Returns (step_fn, items_per_step_per_rank, config_dict)."""
    from sparkdl.models.resnet import ResNet50
    import sparkdl.ops as ops

    torch.manual_seed(1234)
    model = ResNet50().to(device)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)
        batch, hw = args.batch, 224
        opt = ops.FusedSGD(model.parameters(), lr=0.256, momentum=0.875,
                           weight_decay=1 / 32768)
    else:
        batch, hw = 8, 64
        opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import sparkdl.torch as hvd
        hvd.init()
        opt = hvd.DistributedOptimizer(opt)
        hvd.broadcast_parameters(model, root_rank=0)

    x = torch.randn(batch, 3, hw, hw, device=device)
    if use_cuda:
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), device=device)
    autocast_dev = "cuda" if use_cuda else "cpu"

    def step():
        opt.zero_grad()
        with torch.autocast(autocast_dev, dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        return loss

    cfg = {"model": "resnet50", "global_batch": batch * world,
           "seq_len": None, "image": hw, "parallelism": "dp%d" % world}
    return step, batch, cfg


def build_bert_step(args, device, use_cuda):
    from sparkdl.models.bert import BertBase, bert_pretrain_step
    import sparkdl.ops as ops

    torch.manual_seed(1234)
    seq = 512
    batch = args.batch if args.batch != 512 else 64  # per-GPU bert default
    if not use_cuda:
        batch, seq = 2, 128
    model = BertBase().to(device)
    if use_cuda:
        # pure-bf16 weights + fp32 masters in FusedAdamW: no autocast
        # cast storm, bf16 gradient all-reduce
        ops.convert_bf16_training(model)
        opt = ops.FusedAdamW(model.parameters(), lr=1e-4,
                             weight_decay=0.01)
    else:
        opt = torch.optim.AdamW(model.parameters(), lr=1e-4)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import sparkdl.torch as hvd
        hvd.init()
        opt = hvd.DistributedOptimizer(opt)
        hvd.broadcast_parameters(model, root_rank=0)

    step = bert_pretrain_step(model, opt, batch, seq, device, use_cuda)
    cfg = {"model": "bert-base", "global_batch": batch * world,
           "seq_len": seq, "parallelism": "dp%d" % world}
    return step, batch, cfg


def build_gbt_step(args, device, use_cuda):
    """BASELINE config 5: one 'step' = one boosting round on a fixed
    synthetic tabular matrix (HIP histogram kernel when on GPU)."""
    import numpy as np
    from sparkdl.xgboost import gbt

    rng = np.random.RandomState(0)
    # VERDICT round-1 item 6 sizing: 2M x 64 on GPU
    n, f = (2_000_000, 64) if use_cuda else (40_000, 16)
    n = int(os.environ.get("SPARKDL_GBT_ROWS", n))
    X = rng.rand(n, f)
    y = (X[:, 0] * 3 - X[:, 1] ** 2 + 0.3 * rng.randn(n))

    binner = gbt.Binner().fit(X)
    B = binner.transform(X)
    builder = (gbt.GpuHistogramBuilder(B) if use_cuda
               else gbt.CpuHistogramBuilder(B))
    booster = gbt.Booster("reg:squarederror", 0.5, binner, [], f)
    margin = np.full(n, booster._base_margin())
    state = {"margin": margin}

    def step():
        g = state["margin"] - y
        h = np.ones(n)
        pred_out = []
        tree = gbt._build_tree(B, g, h, builder, 6, 1.0, 0.0, 1.0, 0.3,
                               None, pred_out=pred_out)
        booster.trees.append(tree)
        state["margin"] = state["margin"] + (
            pred_out[0] if pred_out else tree.predict_binned(B))
        return len(booster.trees)

    cfg = {"model": "gbt", "rows": n, "features": f, "max_depth": 6,
           "parallelism": "single", "global_batch": n, "seq_len": None}
    return step, 1, cfg


def main():
    args = parse_args()
    use_cuda = torch.cuda.is_available()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))

    if use_cuda:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
        device = torch.device("cuda")
        # MIOpen exhaustive find (SPARKDL_CONV_FIND=0 disables — immediate
        # mode keeps rocprof stats free of autotuning noise)
        torch.backends.cudnn.benchmark = \
            os.environ.get("SPARKDL_CONV_FIND", "1") == "1"
        _load_tunableop()
    else:
        device = torch.device("cpu")

    if args.bucket_mb:
        os.environ["SPARKDL_BUCKET_MB"] = str(args.bucket_mb)

    if world > 1:
        import sparkdl.torch as hvd
        hvd.init()

    if args.allreduce_bench:
        # Bus-bandwidth sweep (SURVEY.md §2.5 scaling-efficiency
        # metric): run under torchrun with N ranks; rank 0 prints one
        # JSON line per size plus a summary line.
        from sparkdl.utils.profiling import CommTimer
        assert world > 1, "--allreduce-bench needs WORLD_SIZE > 1"
        timer = CommTimer()
        best = None
        for mb in (1, 2, 4, 8, 16, 32, 64, 102, 128, 256):
            t = torch.randn(mb * 1024 * 1024 // 4, device=device)
            rec = timer.allreduce(t, iters=20, warmup=5)
            if rank == 0:
                print(json.dumps({"MB": mb, **{
                    k: round(v, 3) if isinstance(v, float) else v
                    for k, v in rec.items()}}), flush=True)
            if best is None or rec["bus_GBps"] > best["bus_GBps"]:
                best = {"MB": mb, **rec}
        if rank == 0:
            print(json.dumps({
                "metric": "allreduce bus bandwidth",
                "value": round(best["bus_GBps"], 1), "unit": "GB/s",
                "n_gpus": world if use_cuda else 0,
                "best_size_MB": best["MB"],
                "roofline_frac": round(best["roofline_frac"], 3),
                "higher_is_better": True}), flush=True)
        import sparkdl.torch as hvd
        hvd.shutdown()
        return

    if args.model == "resnet50":
        step, batch, cfg = build_resnet_step(args, device, use_cuda)
        metric, unit = "images/sec", "images/s"
    elif args.model == "bert":
        step, batch, cfg = build_bert_step(args, device, use_cuda)
        metric, unit = "sequences/sec", "sequences/s"
    else:
        assert world == 1, "gbt bench is single-process"
        step, batch, cfg = build_gbt_step(args, device, use_cuda)
        metric, unit = "boost_rounds/sec", "rounds/s"

    for _ in range(args.warmup):
        step()

    # hipGraph capture of the full train step (fwd+bwd+collectives+opt):
    # launch-bound inner loops replay as one graph. Warmup above has
    # materialized grads, optimizer state, and the multi-tensor tables, so
    # capture sees stable pointers. Falls back to eager on any failure.
    # Graphs measured ~neutral vs eager on these step shapes (launch
    # overhead hides under compute); keep them for the single-GPU path
    # and run multi-rank eager so the round-end scaling sweep cannot be
    # taken down by a capture-time RCCL deadlock.
    if use_cuda and not args.no_graph and world == 1 \
            and args.model != "gbt":  # gbt steps are host-driven
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                step()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                step()
            step = graph.replay
            log("hipGraph capture: ON")
        except Exception as e:  # pragma: no cover
            log("hipGraph capture failed (%s); running eager" % (e,))

    def sync():
        if world > 1:
            import sparkdl.torch as hvd
            hvd.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0

    # MAX elapsed over ranks
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    value = batch * world * args.steps / elapsed
    if rank == 0:
        out = {
            "metric": metric,
            "value": round(value, 2),
            "unit": unit,
            "n_gpus": world if use_cuda else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp64" if args.model == "gbt" else "bf16",
            "data": "synthetic",
            "config": cfg,
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        import sparkdl.torch as hvd
        hvd.shutdown()


if __name__ == "__main__":
    main()
