#!/usr/bin/env python3
# -*- coding: utf-8 -*-
"""Microbenchmark: in-house MFMA flash attention vs AOTriton SDPA.

Times fwd and fwd+bwd on the bench model shapes (GPT-2-medium and
Llama-3-8B) so the STOKE_USE_FA default is set from measurement, not hope.
Run on the GPU box:  python benchmarks/fa_bench.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def bench(fn, warmup=10, iters=50):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def run(tag, B, H, HKV, S, D):
    from stoke.nn.attention import flash_attention

    q = torch.randn(B, H, S, D, device="cuda").bfloat16()
    k = torch.randn(B, HKV, S, D, device="cuda").bfloat16()
    v = torch.randn(B, HKV, S, D, device="cuda").bfloat16()

    def sdpa_f():
        return F.scaled_dot_product_attention(
            q, k, v, is_causal=True, enable_gqa=(H != HKV))

    def fa_f():
        return flash_attention(q, k, v, causal=True)

    t_sdpa_f = bench(sdpa_f)
    t_fa_f = bench(fa_f)

    qg = q.clone().requires_grad_(True)
    kg = k.clone().requires_grad_(True)
    vg = v.clone().requires_grad_(True)
    dout = torch.randn(B, H, S, D, device="cuda").bfloat16()

    def sdpa_fb():
        out = F.scaled_dot_product_attention(
            qg, kg, vg, is_causal=True, enable_gqa=(H != HKV))
        out.backward(dout)
        qg.grad = kg.grad = vg.grad = None

    def fa_fb():
        out = flash_attention(qg, kg, vg, causal=True)
        out.backward(dout)
        qg.grad = kg.grad = vg.grad = None

    t_sdpa_fb = bench(sdpa_fb)
    t_fa_fb = bench(fa_fb)
    # rough flops: fwd 4*B*H*S^2*D/2 (causal), bwd ~2.5x fwd
    fl = 4 * B * H * S * S * D / 2
    print(f"{tag}: fwd sdpa {t_sdpa_f:.3f} ms ({fl/t_sdpa_f/1e9:.0f} TF/s) "
          f"| fa {t_fa_f:.3f} ms ({fl/t_fa_f/1e9:.0f} TF/s)")
    print(f"{tag}: f+b sdpa {t_sdpa_fb:.3f} ms | fa {t_fa_fb:.3f} ms "
          f"| speedup {t_sdpa_fb/t_fa_fb:.2f}x")


if __name__ == "__main__":
    run("gpt2-med B16 H16 S1024 D64 ", 16, 16, 16, 1024, 64)
    run("llama8b  B2  H32/8 S4096 D128", 2, 32, 8, 4096, 128)
