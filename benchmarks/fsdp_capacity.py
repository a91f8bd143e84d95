#!/usr/bin/env python3
# -*- coding: utf-8 -*-
"""FSDP trainable-parameter capacity probe (BASELINE.json metric half 2:
"max FSDP-trainable params/GPU").

Builds a Llama-shaped decoder scaled to --params billions, wraps it in the
in-house FSDP engine (ZeRO-3), runs a few real optimizer steps (bf16 compute,
fp32 flat shards + fp32 Adam state, activation checkpointing), and reports
peak HBM.  Memory model per GPU at world size W:

    fp32 shard 4N/W + fp32 shard grads 4N/W + Adam state 8N/W
    + largest-unit bf16 transient 2N_u + bf16 activations (checkpointed)
    = 16N/W + transients

so a single MI355X (288 GB HBM3E) holds ~15B trainable params without any
host offload (FusedAdamW offload_state moves 8N/W to pinned host for more),
and an 8-GPU node scales toward ~130B (16N/8 = 2N bytes per GPU).

Usage:  torchrun --standalone --nproc-per-node 1 benchmarks/fsdp_capacity.py \
            --params 15 --seq 1024 --steps 2
"""

import argparse
import json
import os
import time


def llama_shape_for(billions: float):
    """Pick (d, nlayer, ffn) giving roughly `billions` params, Llama-style."""
    import math

    vocab = 32000  # smaller vocab: probe measures trainable bulk, not embeds
    # params ~= 2*vocab*d + nlayer * (4*d*d*(nh+2*nkv)/nh/4 ... approximate
    # with standard llama ratios: attn 2.25*d^2 (GQA 8/32), mlp 3*d*ffn,
    # ffn = 3.5*d  ->  per-layer ~= 2.25*d^2 + 10.5*d^2 = 12.75*d^2
    # keep nlayer ~ d/128 (llama-ish aspect)
    target = billions * 1e9
    d = 2048
    while True:
        nlayer = max(8, int(d / 128) * 2)
        per_layer = 12.75 * d * d
        total = 2 * vocab * d + nlayer * per_layer
        if total >= target or d >= 16384:
            break
        d += 1024  # keeps nh = d/128 a multiple of 8 (GQA nkv divides nh)
    # trim depth so the final count lands on (not far above) the target
    per_layer = 12.75 * d * d
    nlayer = max(8, round((target - 2 * vocab * d) / per_layer))
    ffn = int(3.5 * d / 256) * 256
    return vocab, d, nlayer, ffn


def main():
    import torch
    import torch.distributed as dist

    p = argparse.ArgumentParser()
    p.add_argument("--params", type=float, default=15.0, help="billions")
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--steps", type=int, default=2)
    args = p.parse_args()

    from benchmarks.models import Llama
    from stoke.comm import StokeProcessGroup
    from stoke.nn import apply_activation_checkpointing
    from stoke.ops.fused_adam import FusedAdamW
    from stoke.shard import StokeFSDPModule

    rank = int(os.environ.get("RANK", "0"))
    pg = StokeProcessGroup(backend="nccl", init_method="env://",
                           local_rank=int(os.environ.get("LOCAL_RANK", "0")))
    vocab, d, nlayer, ffn = llama_shape_for(args.params)
    nh = max(8, d // 128)
    nkv = max(2, nh // 4)
    if rank == 0:
        print(f"shape: vocab={vocab} d={d} nlayer={nlayer} nh={nh} "
              f"nkv={nkv} ffn={ffn}", flush=True)
    with torch.device("meta"):
        meta = Llama(vocab=vocab, d=d, nlayer=nlayer, nh=nh, nkv=nkv,
                     ffn=ffn, max_seq=args.seq)
    nparams = sum(x.numel() for x in meta.parameters())
    del meta
    if rank == 0:
        print(f"params: {nparams/1e9:.2f}B -> fp32 shard+grad+Adam "
              f"{(16*nparams/pg.world_size)/2**30:.0f} GiB/GPU", flush=True)
    # Materialize directly on the GPU (no 4N-byte host-RAM detour)
    torch.manual_seed(0)
    with torch.device(pg.device):
        model = Llama(vocab=vocab, d=d, nlayer=nlayer, nh=nh, nkv=nkv,
                      ffn=ffn, max_seq=args.seq)
    apply_activation_checkpointing(model)
    fsdp = StokeFSDPModule(model, pg=pg, mixed_precision=True,
                           reshard_after_forward=True)
    opt = FusedAdamW(fsdp.parameters(), lr=1e-4)
    x = torch.randint(0, vocab, (args.batch, args.seq), device=pg.device)
    y = torch.randint(0, vocab, (args.batch, args.seq), device=pg.device)
    torch.cuda.reset_peak_memory_stats()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = fsdp(x)
        loss = torch.nn.functional.cross_entropy(
            out.reshape(-1, vocab).float(), y.reshape(-1)
        )
        loss.backward()
        fsdp.finish_backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    peak = torch.cuda.max_memory_allocated() / 2**30
    if rank == 0:
        print(json.dumps({
            "metric": "max FSDP-trainable params/GPU (probe)",
            "trainable_params_b": round(nparams / 1e9, 2),
            "world_size": pg.world_size,
            "peak_hbm_gib": round(peak, 1),
            "hbm_per_gpu_gib": 288,
            "sec_per_step": round(dt, 2),
            "seq_len": args.seq,
            "batch": args.batch,
            "loss": round(float(loss.detach()), 3),
        }), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    main()
