"""Flash-attention validation/bench harness (docs/FA_PLAN.md step 1).

Compares an implementation (torch SDPA today; axon's CDNA4 kernel in
round 2 via --impl axon) against an fp64 reference on the inputs that
matter: bounded random, spiked-K (forces the online-softmax rescale
branch), and tile-ragged sequence lengths. Runs on CPU (math path) or
GPU.

    python tools/fa_probe.py                    # correctness sweep
    python tools/fa_probe.py --bench --seq 4096 # timing (GPU)
"""
import argparse
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def ref_attention_fp64(q, k, v, causal=True):
    """fp64 reference with GQA expansion; q/k/v: (b, h, s, d)."""
    qf, kf, vf = q.double(), k.double(), v.double()
    hq, hk = qf.shape[1], kf.shape[1]
    if hq != hk:
        kf = kf.repeat_interleave(hq // hk, dim=1)
        vf = vf.repeat_interleave(hq // hk, dim=1)
    s = qf @ kf.transpose(-1, -2) / math.sqrt(qf.shape[-1])
    if causal:
        n = s.shape[-1]
        mask = torch.triu(torch.ones(n, n, dtype=torch.bool,
                                     device=s.device), 1)
        s = s.masked_fill(mask, float("-inf"))
    return torch.softmax(s, dim=-1) @ vf


def run_impl(impl, q, k, v, causal=True):
    if impl == "sdpa":
        return F.scaled_dot_product_attention(q, k, v, is_causal=causal,
                                              enable_gqa=True)
    if impl == "axon":
        from axon import _core  # experimental fwd-only kernel (FA_PLAN v0)

        with torch.no_grad():
            o, _lse = _core.fa_fwd(q, k, v, causal)
        return o
    raise ValueError(impl)


def make_case(kind, b, hq, hk, s, d, device, dtype, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    q = torch.randn(b, hq, s, d, generator=g).clamp(-2, 2)
    k = torch.randn(b, hk, s, d, generator=g).clamp(-2, 2)
    v = torch.randn(b, hk, s, d, generator=g).clamp(-2, 2)
    if kind == "spiked":
        # one huge key vs one query head-row: forces a late max jump (the
        # defer-max rescale branch in an online-softmax kernel)
        k[:, :, s // 2, :] = 6.0
        q[:, :, 3 * s // 4, :] = 6.0
    return (t.to(device, dtype) for t in (q, k, v))


def correctness(args):
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    tol = 3e-2 if dtype == torch.bfloat16 else 2e-3
    worst = 0.0
    for kind in ("random", "spiked"):
        for s in (64, 192, args.seq):  # 192: not a multiple of 64-ish tiles
            q, k, v = make_case(kind, 2, args.heads, args.kv_heads, s,
                                args.head_dim, device, dtype)
            out = run_impl(args.impl, q, k, v)
            ref = ref_attention_fp64(q, k, v)
            err = (out.double() - ref).abs().max().item()
            worst = max(worst, err)
            status = "OK " if err < tol else "FAIL"
            print(f"  [{status}] {kind:7s} seq={s:5d} max|err|={err:.2e}")
            if err >= tol:
                return 1
    print(f"correctness sweep passed (worst {worst:.2e}, tol {tol})")
    return 0


def check_bwd(args):
    """Round-2: validate _core.fa_bwd against autograd through the fp64
    reference (needs a GPU). Error metric is max|err| scaled by
    max(1, max|ref|): dv columns of a strongly-attended key accumulate
    unnormalized sums ≫ 1, so raw absolute error just tracks magnitude.
    SDPA's own backward is scored identically as the production yardstick."""
    assert torch.cuda.is_available(), "--check-bwd needs a GPU"
    from axon import _core

    device, dtype = "cuda", torch.bfloat16
    tol = 3e-2
    ok = True
    for kind in ("random", "spiked"):
        for s in (64, 192, args.seq):
            q, k, v = make_case(kind, 2, args.heads, args.kv_heads, s,
                                args.head_dim, device, dtype)
            g = torch.randn(2, args.heads, s, args.head_dim,
                            device=device).to(dtype)
            q64 = q.double().requires_grad_(True)
            k64 = k.double().requires_grad_(True)
            v64 = v.double().requires_grad_(True)
            ref_attention_fp64(q64, k64, v64).backward(g.double())
            with torch.no_grad():
                o, lse = _core.fa_fwd(q, k, v, True)
                dq, dk, dv = _core.fa_bwd(q, k, v, o, g, lse, True)
            qs = q.detach().requires_grad_(True)
            ks = k.detach().requires_grad_(True)
            vs = v.detach().requires_grad_(True)
            F.scaled_dot_product_attention(qs, ks, vs, is_causal=True,
                                           enable_gqa=True).backward(g)
            for name, got, sdpa_got, want in (
                    ("dq", dq, qs.grad, q64.grad),
                    ("dk", dk, ks.grad, k64.grad),
                    ("dv", dv, vs.grad, v64.grad)):
                norm = max(1.0, want.abs().max().item())
                err = (got.double() - want).abs().max().item() / norm
                sdpa_err = (sdpa_got.double() - want).abs().max().item() / norm
                status = "OK " if err < tol else "FAIL"
                ok &= err < tol
                print(f"  [{status}] {kind:7s} seq={s:5d} {name} "
                      f"err={err:.2e} (sdpa {sdpa_err:.2e})")
    print("bwd check", "passed" if ok else "FAILED")
    return 0 if ok else 1


def bench_bwd(args):
    """Time the axon backward (delta+dkdv+dq kernels) vs SDPA's backward,
    both isolated from their forwards."""
    assert torch.cuda.is_available(), "--bench-bwd needs a GPU"
    import time

    from axon import _core

    device, dtype = "cuda", torch.bfloat16
    q, k, v = make_case("random", args.batch, args.heads, args.kv_heads,
                        args.seq, args.head_dim, device, dtype)
    g = torch.randn(args.batch, args.heads, args.seq, args.head_dim,
                    device=device).to(dtype)
    with torch.no_grad():
        o, lse = _core.fa_fwd(q, k, v, True)
    for _ in range(3):
        with torch.no_grad():
            _core.fa_bwd(q, k, v, o, g, lse, True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        with torch.no_grad():
            _core.fa_bwd(q, k, v, o, g, lse, True)
    torch.cuda.synchronize()
    dt_ax = (time.perf_counter() - t0) / args.steps

    qs = q.detach().requires_grad_(True)
    ks = k.detach().requires_grad_(True)
    vs = v.detach().requires_grad_(True)
    out = F.scaled_dot_product_attention(qs, ks, vs, is_causal=True,
                                         enable_gqa=True)
    for _ in range(3):
        torch.autograd.grad(out, (qs, ks, vs), g, retain_graph=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        torch.autograd.grad(out, (qs, ks, vs), g, retain_graph=True)
    torch.cuda.synchronize()
    dt_sd = (time.perf_counter() - t0) / args.steps

    # bwd FLOPs ~ 2.5x fwd's 2*2*b*h*s^2*d/2 (causal)
    flops = 2.5 * 2 * 2 * args.batch * args.heads * args.seq ** 2 \
        * args.head_dim / 2
    print(f"axon bwd: {dt_ax * 1e3:.2f} ms  ~{flops / dt_ax / 1e12:.0f} TF | "
          f"sdpa bwd: {dt_sd * 1e3:.2f} ms  ~{flops / dt_sd / 1e12:.0f} TF")


def bench(args):
    assert torch.cuda.is_available(), "--bench needs a GPU"
    device, dtype = "cuda", torch.bfloat16
    q, k, v = make_case("random", args.batch, args.heads, args.kv_heads,
                        args.seq, args.head_dim, device, dtype)
    fwd_only = args.impl == "axon" or args.fwd_only
    if not fwd_only:
        q = q.requires_grad_(True)
    for _ in range(3):  # warmup
        out = run_impl(args.impl, q, k, v)
        if not fwd_only:
            out.sum().backward()
            q.grad = None
    torch.cuda.synchronize()
    import time

    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = run_impl(args.impl, q, k, v)
        if not fwd_only:
            out.sum().backward()
            q.grad = None
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    flops = 2 * 2 * args.batch * args.heads * args.seq ** 2 * args.head_dim / 2
    if not fwd_only:
        flops *= 3.5  # fwd+bwd ~ 3.5x fwd (causal already halved)
    tag = "fwd" if fwd_only else "fwd+bwd"
    print(f"{args.impl}: {dt * 1e3:.2f} ms/iter {tag}, "
          f"~{flops / dt / 1e12:.0f} TF effective")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--impl", default="sdpa", choices=["sdpa", "axon"])
    p.add_argument("--seq", type=int, default=512)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--heads", type=int, default=8)
    p.add_argument("--kv-heads", type=int, default=2)
    p.add_argument("--head-dim", type=int, default=128)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--bench", action="store_true")
    p.add_argument("--fwd-only", action="store_true")
    p.add_argument("--bench-bwd", action="store_true")
    p.add_argument("--check-bwd", action="store_true")
    args = p.parse_args()
    if args.check_bwd:
        sys.exit(check_bwd(args))
    if args.bench_bwd:
        sys.exit(bench_bwd(args))
    sys.exit(bench(args) if args.bench else correctness(args))


if __name__ == "__main__":
    main()
