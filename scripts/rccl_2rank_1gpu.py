#!/usr/bin/env python3
"""2-process multi-rank validation on ONE GPU.

Phase 1 — try the real nccl(=RCCL) backend with both ranks pinned to
cuda:0. RCCL refuses duplicate devices in one communicator ("Duplicate GPU
detected"); the refusal is recorded as the hardware's answer (a true RCCL
all_to_all_single needs >=2 GPUs, which the driver's round-end scale run
provides).

Phase 2 — run the full multi-rank GPU engine (padded route: HIP
k_bucketize_pad / owner unique+gather / k_scatter_out / k_gather_pad /
k_split_payload, handle bookkeeping, dense allreduce ordering) across two
processes sharing cuda:0, with gloo as the wire transport (CPU staging in
parallel/comm.py). Everything except the literal RCCL collective call is
the production multi-rank path. Validates the final table state against a
world-1 reference over the union of both ranks' batches.

    python scripts/rccl_2rank_1gpu.py
    (writes gpurun_out/rccl_2rank.log)
"""

import json
import os
import socket
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

DIM = 9
VOCAB = 100_000
STEPS = 4
N = 4096


def _batches(rank):
    import torch
    g = torch.Generator().manual_seed(1000 + rank)
    out = []
    for _ in range(STEPS):
        keys = torch.randint(0, VOCAB, (N,), dtype=torch.int64, generator=g)
        grads = torch.randn(N, DIM, generator=g)
        out.append((keys, grads))
    return out


def _make_var(ctx):
    st = ctx.create_storage()
    var = st.create_variable(VOCAB, DIM)
    var.set_initializer("uniform", minval=-1.0, maxval=1.0)
    var.set_optimizer("adagrad", learning_rate=0.05,
                      initial_accumulator_value=0.1, epsilon=1e-10)
    return st, var


def worker(backend):
    import torch
    import torch.distributed as dist
    from openembedding_amd.context import Context

    rank = int(os.environ["RANK"])
    dist.init_process_group(backend)
    torch.cuda.set_device(0)
    ctx = Context(device="cuda:0")
    st, var = _make_var(ctx)

    t0 = time.time()
    for keys, grads in _batches(rank):
        out, h = var.pull(keys.cuda())
        var.push(h, grads.cuda())
        st.update_weights()
    torch.cuda.synchronize()
    dist.barrier()
    elapsed = time.time() - t0
    var.check_padded_overflow()

    # dense-allreduce sanity over the same transport
    t = torch.full((1024,), float(rank + 1))
    t = t.cuda() if backend == "nccl" else t
    dist.all_reduce(t)
    assert torch.all(t == 3.0), "allreduce mismatch"

    probe = torch.unique(torch.cat(
        [k for r in range(2) for k, _ in _batches(r)])).cuda()
    after, _ = var.pull(probe, readonly=True)
    padded = var._use_padded()
    dist.destroy_process_group()
    if rank != 0:
        return
    # world-1 reference over the union, interleaved in step order
    os.environ["WORLD_SIZE"] = "1"
    ref_ctx = Context(device="cuda:0")
    ref_st, ref = _make_var(ref_ctx)
    batches = [_batches(0), _batches(1)]
    for t_i in range(STEPS):
        hs = []
        for r in range(2):
            keys, grads = batches[r][t_i]
            _, h = ref.pull(keys.cuda())
            hs.append((h, grads))
        for h, grads in hs:
            ref.push(h, grads.cuda())
        ref_st.update_weights()
    ref_after, _ = ref.pull(probe, readonly=True)
    import torch as _t
    _t.testing.assert_close(after, ref_after, rtol=1e-4, atol=1e-5)
    info = {"ok": True, "backend": backend, "padded_mode": padded,
            "steps": STEPS, "elapsed_s": elapsed,
            "probe_keys": int(probe.numel())}
    print("TWORANK_OK " + json.dumps(info), flush=True)


def _spawn(backend, timeout_s):
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update(RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK="0",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        procs.append(subprocess.Popen(
            [sys.executable, os.path.abspath(__file__), "--worker", backend],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=timeout_s)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
        outs.append(out.decode())
    ok = all(p.returncode == 0 for p in procs) and "TWORANK_OK" in outs[0]
    return ok, "\n--- rank1 ---\n".join(outs)


def main():
    log = []
    log.append("== phase 1: nccl (RCCL), both ranks on cuda:0 ==")
    ok_nccl, text = _spawn("nccl", 240)
    log.append(text)
    log.append(f"nccl 2-rank-1-GPU: {'PASS' if ok_nccl else 'REFUSED'}")
    log.append("")
    log.append("== phase 2: full GPU engine, gloo transport ==")
    ok_gloo, text = _spawn("gloo", 420)
    log.append(text)
    log.append(f"gloo-transport GPU engine 2-rank: "
               f"{'PASS' if ok_gloo else 'FAIL'}")
    body = "\n".join(log)
    os.makedirs(os.path.join(ROOT, "gpurun_out"), exist_ok=True)
    with open(os.path.join(ROOT, "gpurun_out", "rccl_2rank.log"), "w") as f:
        f.write(body)
    print(body[-3000:])
    # phase 2 must pass; phase 1 passing would be a bonus
    sys.exit(0 if ok_gloo else 1)


if __name__ == "__main__":
    if "--worker" in sys.argv:
        worker(sys.argv[sys.argv.index("--worker") + 1])
    else:
        main()
