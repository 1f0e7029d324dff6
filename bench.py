#!/usr/bin/env python3
"""Flagship benchmark: DeepFM on Criteo-shaped synthetic data.

Measures the BASELINE.json metric — samples/sec for DeepFM (embedding dim 9,
Adagrad, batch 4096 per GPU, Criteo Kaggle field cardinalities) — on
1..8 MI355X GPUs, one rank per GPU over RCCL.

    python bench.py --gpus N --steps K --warmup W
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N ...

Rank 0 prints one JSON line; at the default --amp native it carries BOTH the
bf16-resident-MLP headline and a matched-precision all-fp32 number
("matched_precision", measured back-to-back in the same process).
Baseline: the reference's published DeepFM dim9 throughput on 8x Tesla T4
(BASELINE.md): 293/458/727/935 k samples/s at 1/2/4/8 GPUs.

Multi-rank: the embedding all-to-all runs the padded sync-free route
(parallel/sharded.py _pull_remote_padded) so the whole step — collectives
included — is hipGraph-capturable; if capture is unavailable the bench
falls back to plain eager (the pulling() prefetch pipeline is opt-in via
--prefetch: measured net-negative against the ~20 us xGMI wire).
"""

import argparse
import json
import os
import time

# hipBLASLt algo autotuning during the (untimed) warmup: measured +8% on the
# DeepFM step (gpurun_out/bench7_tunable.log vs bench7.log). Must be set
# before torch initializes its blas handles. Pre-tuned MI355X results ship
# in profiles/ so steady runs skip most of the tuning; shapes not in the
# file still tune on first use.
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
_tuned = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "profiles", "tunableop_mi355x_%d.csv")
if os.path.exists(_tuned % 0):
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _tuned)

import torch
import torch.distributed as dist

# reference OE DeepFM dim9 samples/s at 1/2/4/8 GPUs
BASELINE_SAMPLES_PER_SEC = {1: 293_000.0, 2: 458_000.0, 4: 727_000.0,
                            8: 935_000.0}


def build_model(args, embed, amp_mode, device, on_gpu):
    from openembedding_amd.models import MODELS

    torch.manual_seed(1234)  # identical dense init on all ranks
    kw = {} if args.model == "lr" else {"dim": args.dim}
    if args.hash and args.model != "lr":
        kw["hash_mode"] = True
    model = MODELS[args.model](**kw).to(device)
    if hasattr(model, "head_bf16"):
        model.head_bf16 = amp_mode == "bf16"  # deep_in dtype follows amp
    if amp_mode == "native" and on_gpu:
        from openembedding_amd.models.ctr import convert_mlp_bf16
        convert_mlp_bf16(model)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.005),
        flatten_dense=True)
    return model, opt


def measure(args, embed, amp_mode, steps, warmup, pool, device, world, rank):
    """Build the model at ``amp_mode`` and time ``steps`` steps. Returns a
    dict with value/ms_per_step/graph/prefetch."""
    on_gpu = device.type == "cuda"
    model, opt = build_model(args, embed, amp_mode, device, on_gpu)
    # fused 2-kernel BCE on GPU (torch's spends ~5 launches/step in the
    # captured graph), plain torch BCE elsewhere
    from openembedding_amd.ops.dispatch import bce_with_logits as lossf

    amp = amp_mode == "bf16" and on_gpu
    # cached backward seed: loss.backward() re-fills a ones scalar every
    # step (~4 us launch in the captured graph)
    seed_one = torch.ones((), device=device) if on_gpu else None

    def run_step(dense, sparse, labels):
        opt.zero_grad(set_to_none=False)
        # cache_enabled=False is required for hipGraph capture under autocast
        # (the autocast weight-cast cache is not capture-safe)
        with torch.autocast("cuda", dtype=torch.bfloat16, enabled=amp,
                            cache_enabled=False):
            out = model(dense, sparse)
        loss = lossf(out.float(), labels)
        loss.backward(gradient=seed_one)
        opt.step()
        return loss

    # hipGraph capture: single-GPU since round 1; multi-rank too now that
    # the padded all-to-all route is sync-free (RCCL collectives capture
    # into the graph; if RCCL capture is unavailable we fall back to eager
    # + prefetch overlap below)
    use_graph = on_gpu and args.graph != "off"
    graph = None
    static = None
    if use_graph:
        static = tuple(t.clone() for t in pool[0])
        try:
            for _ in range(3):  # eager warmup (lazy state, GEMM algo select)
                run_step(*static)
            torch.cuda.synchronize()
            if world > 1:
                dist.barrier()  # all ranks enter capture together
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    run_step(*static)
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                run_step(*static)
        except Exception as e:  # noqa: BLE001
            if args.graph == "on":
                raise
            print(f"[bench] hipGraph capture unavailable ({e!r}); "
                  f"falling back to eager", flush=True)
            graph = None
        if world > 1 and dist.is_initialized():
            # unanimity: a rank whose capture failed must not meet captured
            # ranks' collectives with eager ones — all fall back together
            ok = torch.tensor([1.0 if graph is not None else 0.0],
                              device=device if dist.get_backend() == "nccl"
                              else "cpu")
            dist.all_reduce(ok, op=dist.ReduceOp.MIN)
            if float(ok.item()) == 0.0:
                graph = None

    # prefetch: opt-in only. Measured (gpurun r2ag): the prefetch stream
    # costs ~80 us/step of stream-switch overhead at world 1 — more than
    # the ~20 us padded-wire time it could hide at world>1 (the
    # reference's prefetch paid off against millisecond RPC latencies;
    # xGMI is not that). Graph capture is the multi-rank fast path; plain
    # eager is the fallback.
    want_prefetch = args.prefetch
    if want_prefetch and graph is None:
        def batch_stream(total):
            for i in range(total):
                yield pool[i % len(pool)]

        stream = embed.pulling(batch_stream(warmup + steps), model)

        def step(i):
            dense, sparse, labels = next(stream)
            return run_step(dense, sparse, labels)
    else:
        want_prefetch = False

        def step(i):
            dense, sparse, labels = pool[i % len(pool)]
            if graph is not None:
                static[0].copy_(dense)
                static[1].copy_(sparse)
                static[2].copy_(labels)
                graph.replay()
                return None
            return run_step(dense, sparse, labels)

    for i in range(warmup):
        step(i)

    if dist.is_initialized():
        dist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        step(warmup + i)
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # a padded-wire overflow would mean some keys silently read zeros: the
    # run must fail loudly instead of reporting a wrong-but-fast number
    ctx = embed.get_context()
    for v in ctx.variables.values():
        if hasattr(v, "check_padded_overflow"):
            v.check_padded_overflow()

    # max across ranks = whole-job time
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if dist.is_initialized()
                     and dist.get_backend() == "nccl" else "cpu")
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    n_gpus = world if world > 1 else args.gpus
    return {
        "value": args.batch * n_gpus * steps / elapsed,
        "ms_per_step": elapsed / steps * 1000.0,
        "graph": graph is not None,
        "prefetch": bool(want_prefetch),
        "steps": steps,
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=4096, help="per-GPU batch")
    p.add_argument("--model", default="deepfm",
                   choices=["deepfm", "wdl", "xdeepfm", "lr"])
    p.add_argument("--dim", type=int, default=9)
    p.add_argument("--data-pool", type=int, default=8,
                   help="pre-generated synthetic batches, rotated")
    p.add_argument("--amp", default="native",
                   choices=["bf16", "off", "native"],
                   help="dense-MLP precision: off = fp32; bf16 = autocast "
                        "(per-step weight casts); native (default) = "
                        "bf16-resident MLP weights + the fused single-kernel "
                        "MLP, fp32 master/accumulators in the flat optimizer. "
                        "Embeddings, FM math, loss, optimizer state fp32. "
                        "The default also re-measures all-fp32 and reports "
                        "it as matched_precision in the same JSON line.")
    p.add_argument("--graph", default="auto", choices=["auto", "on", "off"],
                   help="capture the train step in a hipGraph")
    p.add_argument("--hash", action="store_true",
                   help="store embeddings in the open-addressed hash table "
                        "(lazy rows) instead of the array table")
    p.add_argument("--prefetch", action="store_true",
                   help="force the pulling() prefetch pipeline (default: "
                        "automatic at world>1 when the step is not "
                        "graph-captured)")
    p.add_argument("--no-matched-precision", action="store_true",
                   help="skip the secondary all-fp32 measurement")
    p.add_argument("--cache-mb", type=int, default=0,
                   help="capacity tier: device row-cache budget in MB, cold "
                        "rows spill to host DRAM (implies --hash)")
    args = p.parse_args()
    if args.cache_mb:
        args.hash = True
        import openembedding_amd as oe
        oe.flags.config = f"server:\n  cache_size: {args.cache_mb}\n"

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))

    import openembedding_amd.torch as embed
    from openembedding_amd.models import synthetic_batch

    ctx = embed.get_context()
    device = ctx.device
    on_gpu = device.type == "cuda"
    if on_gpu:
        from openembedding_amd.ops import require_hip
        require_hip()  # fail loudly if the native extension is missing

    gen = torch.Generator(device="cpu").manual_seed(4242 + rank)
    pool = []
    for _ in range(args.data_pool):
        dense, sparse, labels = synthetic_batch(args.batch, generator=gen)
        pool.append((dense.to(device), sparse.to(device), labels.to(device)))

    amp_mode = args.amp
    main_res = measure(args, embed, amp_mode, args.steps, args.warmup, pool,
                       device, world, rank)

    matched = None
    if (amp_mode == "native" and on_gpu and not args.no_matched_precision
            and args.model != "lr"):
        msteps = max(10, args.steps // 4)
        mres = measure(args, embed, "off", msteps, max(5, args.warmup // 2),
                       pool, device, world, rank)
        matched = {"dtype": "fp32", "value": mres["value"],
                   "ms_per_step": mres["ms_per_step"],
                   "steps": mres["steps"], "graph": mres["graph"]}

    n_gpus = world if world > 1 else args.gpus
    samples_per_sec = main_res["value"]
    baseline = BASELINE_SAMPLES_PER_SEC.get(n_gpus)
    result = {
        "metric": "samples/sec DeepFM Criteo",
        "value": samples_per_sec,
        "unit": "samples/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": main_res["ms_per_step"],
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": (samples_per_sec / baseline) if (
            baseline and args.model == "deepfm" and args.dim == 9
            and args.batch == 4096) else None,
        "dtype": ("bf16" if (args.amp in ("bf16", "native") and on_gpu)
                  else "fp32"),
        "data": "synthetic Criteo-shaped (random ids, Criteo-Kaggle "
                "cardinalities, random labels), random-init weights",
        "matched_precision": matched,
        "config": {
            "model": f"{args.model}-dim{args.dim}",
            "global_batch": args.batch * n_gpus,
            "fields": 26,
            "dense_features": 13,
            "optimizer": "adagrad",
            "precision_note": (
                "dense MLP bf16 autocast (MFMA); embeddings, FM reductions "
                "fp32; optimizer state fp32" if args.amp == "bf16" and on_gpu
                else
                "dense MLP weights bf16-resident (MFMA) with fp32 master + "
                "fp32 accumulators; embeddings, FM math, loss fp32"
                if (args.amp == "native" and on_gpu) else "all fp32"),
            "graph": main_res["graph"],
            "prefetch": main_res["prefetch"],
            "table": "hash" if args.hash else "array",
            "cache_mb": args.cache_mb,
            "parallelism": (f"dense-dp{n_gpus} + embedding sharded "
                            f"all_to_all" if n_gpus > 1 else "single-gpu"),
        },
    }
    if rank == 0:
        print(json.dumps(result))


if __name__ == "__main__":
    main()
