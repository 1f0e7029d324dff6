"""Attribute the DeepFM step's small-kernel tail to host-side ops.

The rocprofv3 kernel table (profiles/bench_deepfm_1gpu_kernels_r2.md) shows
~4 FillFunctor and ~6 unrolled-elementwise launches per step that the kernel
names alone don't attribute. torch.profiler gives the aten op behind each
launch. Eager (uncaptured) steps — same construction as bench.py's headline
config (DeepFM dim9, batch 4096, native amp = bf16 MLP weights).

Run on a GPU box:
    python scripts/step_attrib.py > gpurun_out/step_attrib.txt
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def main():
    import openembedding_amd.torch as embed
    from openembedding_amd.models import MODELS, synthetic_batch
    from openembedding_amd.models.ctr import convert_mlp_bf16
    from openembedding_amd.ops.dispatch import bce_with_logits as lossf

    dev = "cuda:0"
    torch.manual_seed(1234)
    model = MODELS["deepfm"](dim=9).to(dev)
    convert_mlp_bf16(model)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.005),
        flatten_dense=True)

    gen = torch.Generator().manual_seed(7)
    pool = [tuple(t.to(dev) for t in synthetic_batch(4096, generator=gen))
            for _ in range(4)]

    def step(b):
        opt.zero_grad(set_to_none=False)
        dense, sparse, labels = b
        loss = lossf(model(dense, sparse).float(), labels)
        loss.backward()
        opt.step()

    for i in range(10):
        step(pool[i % 4])
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]
                 ) as prof:
        for i in range(8):
            step(pool[i % 4])
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="self_cuda_time_total",
                                    row_limit=48, max_name_column_width=64))

    # second pass with python stacks: where do the remaining fill_/copy_
    # launches come from?
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 with_stack=True) as prof2:
        for i in range(8):
            step(pool[i % 4])
        torch.cuda.synchronize()
    print("\n==== fill_/copy_/add_ by stack ====")
    tbl = prof2.key_averages(group_by_stack_n=6)
    for e in sorted(tbl, key=lambda e: -e.self_device_time_total):
        if e.key.split(".")[-1] not in ("fill_", "copy_", "add_", "zero_"):
            continue
        print(f"\n{e.key}  n={e.count}  self_cuda={e.self_device_time_total:.0f}us")
        for line in (e.stack or [])[:6]:
            print("   ", line)


if __name__ == "__main__":
    main()
