#!/usr/bin/env python3
# -*- coding: utf-8 -*-
"""Causal-LM training example: Llama-shape decoder under the Stoke API.

Shows the pieces the CIFAR example doesn't: ``BucketedDistributedSampler``
(variable-length sequences bucketed to minimize padding), FSDP (ZeRO-3)
sharding for models past single-GPU memory, activation checkpointing, and
the fused-Adam optimizer — all through the same declarative facade.

Single GPU:
    python train.py --layers 4 --dim 512
8-GPU FSDP:
    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 train.py --fsdp
CPU smoke (tiny shapes):
    python train.py --cpu --layers 2 --dim 256 --steps 4
"""

import argparse
import os
import sys

import torch
import torch.nn as nn

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from benchmarks.models import Llama  # noqa: E402
from stoke import (  # noqa: E402
    BucketedDistributedSampler,
    ClipGradNormConfig,
    DDPConfig,
    Stoke,
    StokeOptimizer,
)
from stoke.nn import apply_activation_checkpointing  # noqa: E402
from stoke.ops.fused_adam import FusedAdamW  # noqa: E402


class SyntheticDocs(torch.utils.data.Dataset):
    """Variable-length token sequences, padded per batch (bucketing keeps
    batches length-homogeneous so the padding is minimal)."""

    def __init__(self, n=2048, vocab=1024, min_len=32, max_len=256):
        g = torch.Generator().manual_seed(0)
        self.lens = torch.randint(min_len, max_len + 1, (n,), generator=g)
        self.vocab = vocab
        self.max_len = max_len

    def __len__(self):
        return len(self.lens)

    def __getitem__(self, i):
        i = int(i)  # sampler indices may be numpy ints
        L = int(self.lens[i])
        g = torch.Generator().manual_seed(1000 + i)
        toks = torch.randint(0, self.vocab, (L,), generator=g)
        return toks

    def sorted_indices(self):
        return torch.argsort(self.lens).tolist()


def pad_collate(batch):
    # inputs pad with 0; TARGETS pad with -100 so padded positions are
    # ignored by the loss instead of trained as real next tokens
    # 
    L = max(t.numel() for t in batch)
    x = torch.zeros(len(batch), L, dtype=torch.long)
    y = torch.full((len(batch), L), -100, dtype=torch.long)
    for i, t in enumerate(batch):
        x[i, : t.numel()] = t
        y[i, : t.numel()] = t
    return x[:, :-1], y[:, 1:]  # next-token prediction


def lm_loss(logits, target):
    from stoke.nn import fused_cross_entropy

    return fused_cross_entropy(logits, target, ignore_index=-100)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cpu", action="store_true")
    ap.add_argument("--fsdp", action="store_true")
    ap.add_argument("--layers", type=int, default=4)
    ap.add_argument("--dim", type=int, default=512)
    ap.add_argument("--vocab", type=int, default=1024)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--buckets", type=int, default=4)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--act-ckpt", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    distributed = world > 1
    nh = max(4, args.dim // 128)
    model = Llama(vocab=args.vocab, d=args.dim, nlayer=args.layers, nh=nh,
                  nkv=max(2, nh // 2), ffn=2 * args.dim, max_seq=512)
    if args.act_ckpt:
        apply_activation_checkpointing(model, min_params=10_000)

    s = Stoke(
        model=model,
        optimizer=StokeOptimizer(
            optimizer=FusedAdamW if not args.cpu else torch.optim.AdamW,
            optimizer_kwargs={"lr": 3e-4},
        ),
        loss=lm_loss,
        batch_size_per_device=args.batch,
        grad_clip=ClipGradNormConfig(max_norm=1.0, norm_type=2.0),
        gpu=not args.cpu,
        fp16=None if args.cpu else "bf16",
        distributed="ddp" if distributed else None,
        fairscale_fsdp=args.fsdp and distributed,
        configs=[DDPConfig(local_rank=int(os.environ.get("LOCAL_RANK", 0)))],
        verbose=False,
    )
    ds = SyntheticDocs(vocab=args.vocab)
    sampler = BucketedDistributedSampler(
        ds,
        buckets=args.buckets,
        batch_size=args.batch,
        sorted_idx=ds.sorted_indices(),
        num_replicas=s.world_size if distributed else 1,
        rank=s.rank if distributed else 0,
        drop_last=True,
    )
    loader = s.DataLoader(
        ds, sampler=sampler if distributed else None,
        collate_fn=pad_collate, drop_last=True,
    )
    s.print_num_model_parameters()
    it = 0
    for epoch in range(100):
        sampler.set_epoch(epoch)
        # non-distributed runs iterate the sampler's bucketed order directly
        idx_iter = iter(sampler) if not distributed else None
        if distributed:
            batches = loader
        else:
            idxs = list(idx_iter)
            batches = (
                pad_collate([ds[j] for j in idxs[k : k + args.batch]])
                for k in range(0, len(idxs) - args.batch + 1, args.batch)
            )
        for x, y in batches:
            if not args.cpu:
                x, y = x.cuda(), y.cuda()
            out = s.model(x)
            loss = s.loss(out, y)
            s.backward(loss)
            s.step()
            it += 1
            if it % 10 == 0:
                s.print_ema_loss()
            if it >= args.steps:
                s.print(f"done: {it} steps, final EMA loss "
                        f"{s.ema_loss:.3f}")
                return


if __name__ == "__main__":
    main()
