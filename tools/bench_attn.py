"""Attention kernel microbenchmark at the Llama-3-8B bench shape.

  python tools/bench_attn.py [--s 4096] [--b 1] [--hq 32] [--hkv 8] [--iters 20]

Prints achieved TFLOP/s for fwd and bwd (causal-halved flop count).
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from megatron_amd import ops


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--s", type=int, default=4096)
    p.add_argument("--b", type=int, default=1)
    p.add_argument("--hq", type=int, default=32)
    p.add_argument("--hkv", type=int, default=8)
    p.add_argument("--d", type=int, default=128)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--no-bwd", action="store_true")
    args = p.parse_args()
    s, b, hq, hkv, d = args.s, args.b, args.hq, args.hkv, args.d

    q = torch.randn(s, b, hq, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(s, b, hkv, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(s, b, hkv, d, device="cuda", dtype=torch.bfloat16, requires_grad=True)

    # causal-halved: fwd = 2 gemms, bwd = 5 gemm-equivalents (2 recomputed)
    fwd_flops = 2 * 2 * (s * s / 2) * d * hq * b
    bwd_flops = fwd_flops / 2 * 7  # ST,dPT,dq + ST,dPT(dkv recompute),dV,dK

    out, lse = ops._C.attn_fwd(q, k, v, True, d**-0.5, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        out, lse = ops._C.attn_fwd(q, k, v, True, d**-0.5, 0)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    print(f"fwd: {dt*1e3:.3f} ms  {fwd_flops/dt/1e12:.1f} TF (useful-flop basis)")

    if not args.no_bwd:
        dy = torch.randn_like(out)
        ops._C.attn_bwd(dy, q, k, v, out, lse, True, d**-0.5, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            ops._C.attn_bwd(dy, q, k, v, out, lse, True, d**-0.5, 0)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        print(f"bwd: {dt*1e3:.3f} ms  {bwd_flops/dt/1e12:.1f} TF (issued-flop basis)")


if __name__ == "__main__":
    main()
