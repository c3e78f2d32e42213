"""Benchmark entry point (driver contract).

Flagship (default, BASELINE.json metric): ResNet-50, synthetic
ImageNet-shaped data, bf16, data-parallel over N MI355X GPUs —
whole-job images/sec.

`python bench.py --gpus N --steps K --warmup W`; for N>1 the driver
launches one rank per GPU via torch.distributed.run. W untimed warmup
steps, then exactly K timed steps bracketed by barrier +
torch.cuda.synchronize on both sides; elapsed = MAX over ranks; rank 0
prints ONE JSON line with the whole-job aggregate.

Secondary configs (BASELINE.json configs 3-5) are reachable with
--model bert-large (LAMB + fp16 compression), --model llama3-8b
(pure-DP bf16 + fused AdamW), --reduction adasum (ResNet-50 Adasum).
"""
import argparse
import json
import os
import time

import torch
import torch.nn.functional as F

import axon as hvd

CONFIGS = os.path.join(os.path.dirname(os.path.abspath(__file__)), "configs")


def _load_tunableop(tag):
    """Apply committed hipBLASLt TunableOp solution tables (tuned once on
    an MI355X, configs/tunableop_*.csv): tuning itself stays off, so
    startup cost is zero and runs are deterministic."""
    path = os.path.join(CONFIGS, f"tunableop_{tag}.csv")
    if not (torch.cuda.is_available() and os.path.exists(path)):
        return
    try:
        t = torch.cuda.tunable
        t.enable(True)
        t.tuning_enable(False)
        t.read_file(path)
    except Exception as e:  # pragma: no cover - best effort
        print(f"[bench] tunableop load failed: {e}", flush=True)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=None, help="per-GPU batch size")
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--seq-len", type=int, default=None)
    p.add_argument("--model", default="resnet50",
                   choices=["resnet50", "bert-large", "llama3-8b", "llama3-1b"])
    p.add_argument("--reduction", default="average", choices=["average", "adasum"])
    p.add_argument("--compression", default=None, choices=[None, "none", "fp16"])
    p.add_argument("--no-fused-optimizer", action="store_true")
    return p.parse_args()


def build_resnet(args, device, dtype):
    from axon.models.resnet import resnet50
    from axon.optim import FusedSGD

    # MIOpen immediate-find: benchmark mode picks the fastest conv algo per
    # shape during warmup (shapes are static here).
    torch.backends.cudnn.benchmark = True
    torch.manual_seed(1234)
    model = resnet50().to(device=device, dtype=dtype)
    if device.type == "cuda":
        # NHWC: MIOpen's MFMA conv kernels want channels-last on CDNA4.
        model = model.to(memory_format=torch.channels_last)
    batch = args.batch or 256
    if args.no_fused_optimizer or device.type == "cpu":
        opt = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9,
                              weight_decay=1e-4)
    else:
        opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                       weight_decay=1e-4)
    g = torch.Generator().manual_seed(hvd.rank())
    x = torch.randn(batch, 3, args.image_size, args.image_size,
                    generator=g).to(device=device, dtype=dtype)
    if device.type == "cuda":
        x = x.to(memory_format=torch.channels_last)
    y = torch.randint(0, 1000, (batch,), generator=g).to(device)

    def fwd():
        return F.cross_entropy(model(x).float(), y)

    def step():
        opt.zero_grad(set_to_none=True)
        loss = fwd()
        loss.backward()
        opt.step()
        return loss

    step.fwd = fwd
    cfg = {"model": "resnet50", "global_batch": batch * hvd.size(),
           "image_size": args.image_size}
    return model, opt, step, batch, "images/sec", cfg


def build_bert(args, device, dtype):
    from axon.models.bert import BertForPreTraining, bert_large_config
    from axon.optim import FusedLAMB

    _load_tunableop("bert")
    torch.manual_seed(1234)
    with torch.device(device):
        model = BertForPreTraining(bert_large_config())
    model = model.to(dtype=dtype)
    batch = args.batch or 64
    seq = args.seq_len or 128
    if args.no_fused_optimizer or device.type == "cpu":
        opt = torch.optim.AdamW(model.parameters(), lr=1e-4, weight_decay=0.01)
    else:
        opt = FusedLAMB(model.parameters(), lr=1e-3, weight_decay=0.01)
    g = torch.Generator().manual_seed(hvd.rank())
    ids = torch.randint(0, 30522, (batch, seq), generator=g).to(device)
    labels = ids.clone()
    mask = torch.rand(batch, seq, generator=g) > 0.15
    labels[mask.to(device)] = -100

    def fwd():
        return model.mlm_loss(ids, labels)

    def step():
        opt.zero_grad(set_to_none=True)
        loss = fwd()
        loss.backward()
        opt.step()
        return loss

    step.fwd = fwd
    cfg = {"model": "bert-large", "global_batch": batch * hvd.size(),
           "seq_len": seq, "optimizer": "fused_lamb"}
    return model, opt, step, batch * seq, "tokens/sec", cfg


def build_llama(args, device, dtype, which="llama3-8b"):
    from axon.models.llama import (LlamaForCausalLM, llama3_1b_config,
                                   llama3_8b_config)
    from axon.optim import FusedAdamW

    if which == "llama3-8b":
        _load_tunableop("llama")
    fa = os.environ.get("AXON_FA_LIB")
    if fa and torch.cuda.is_available():
        # "ck" selects composable_kernel flash attention when torch is
        # built with it (this image's torch is not — AOTriton only).
        try:
            torch.backends.cuda.preferred_rocm_fa_library(fa)
        except RuntimeError as e:
            print(f"[bench] AXON_FA_LIB={fa} unavailable: {e}", flush=True)
    torch.manual_seed(1234)
    cfgm = llama3_8b_config() if which == "llama3-8b" else llama3_1b_config()
    # Construct directly on the device: a CPU-first build of 8B params
    # costs 32 GB host RAM and minutes of single-threaded init PER RANK
    # (x8 on the scaling run); on-device init is seconds.
    with torch.device(device):
        model = LlamaForCausalLM(cfgm)
    model = model.to(dtype=dtype)
    batch = args.batch or (8 if which == "llama3-8b" else 8)
    seq = args.seq_len or 4096
    if args.no_fused_optimizer or device.type == "cpu":
        opt = torch.optim.AdamW(model.parameters(), lr=3e-4)
    else:
        opt = FusedAdamW(model.parameters(), lr=3e-4)
    g = torch.Generator().manual_seed(hvd.rank())
    ids = torch.randint(0, cfgm.vocab_size, (batch, seq), generator=g).to(device)

    def fwd():
        return model.lm_loss(ids)

    def step():
        opt.zero_grad(set_to_none=True)
        loss = fwd()
        loss.backward()
        opt.step()
        return loss

    step.fwd = fwd
    cfg = {"model": which, "global_batch": batch * hvd.size(), "seq_len": seq,
           "optimizer": "fused_adamw"}
    return model, opt, step, batch * seq, "tokens/sec", cfg


def try_graph_step(opt, fwd, n, op, compression):
    """Capture forward+loss+backward of one training step in a hipGraph and
    return a replay-based step(), or None if capture is unavailable.

    MI355X rationale: the ResNet-50 step is ~700 small kernel launches in
    ~30 ms and rocprof shows ~11% of GPU idle gaps between them; replaying
    one captured graph removes the per-launch host overhead. The optimizer
    and the gradient allreduce stay OUTSIDE the graph: the engine's
    background RCCL cycle (host-thread driven) is not capturable, and the
    fused-optimizer launcher stages a transient pinned chunk table whose
    captured H2D copy would dangle on replay. The captured backward's
    grad outputs live in the graph's private pool with stable addresses,
    so the post-replay grouped allreduce + fused step reuse them in place
    (never call zero_grad between replays — replayed producers overwrite
    the same buffers).
    """
    try:
        # Per-parameter overlap hooks would fire at capture time (host
        # callbacks don't replay) and enqueue engine work on uncaptured
        # streams — graph mode allreduces the stable grad buffers instead.
        # Removed before the side-stream warmup so no collective is in
        # flight when capture begins; main() re-registers them if any rank
        # fails to capture (all ranks must agree on a mode or the
        # negotiation name sets diverge and the job stalls).
        for h in getattr(opt, "_hook_handles", []):
            h.remove()
        if hasattr(opt, "_hook_handles"):
            opt._hook_handles = []
        params = [p for group in opt.param_groups
                  for p in group["params"] if p.requires_grad]
        # The backward is captured as torch.autograd.grad on the CALLING
        # thread (multithreading off). Both choices are load-bearing on
        # ROCm, measured on MI355X:
        #  - loss.backward() routes through AccumulateGrad nodes; a node
        #    cached from an earlier eager iteration is pinned to another
        #    stream, and its cross-stream hop inside capture either
        #    escapes capture silently (thread_local mode -> replay applied
        #    stale grads) or kills the capture (global mode -> segfault).
        #    autograd.grad has no AccumulateGrad nodes at all.
        #  - the default multithreaded engine runs backward on a worker
        #    thread, racing the capture-mode bookkeeping; single-threaded
        #    execution keeps every launch on the capturing thread.
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side), \
                torch.autograd.set_multithreading_enabled(False):
            for _ in range(2):
                loss = fwd()
                torch.autograd.grad(loss, params)
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g), \
                torch.autograd.set_multithreading_enabled(False):
            static_loss = fwd()
            grads = list(torch.autograd.grad(static_loss, params))
        if not grads or any(gr is None for gr in grads):
            raise RuntimeError("capture produced no gradients")
        # Bind the captured (graph-pool, stable-address) grad buffers to
        # .grad once; every replay rewrites them in place and the fused
        # optimizer reads them between replays.
        opt.zero_grad(set_to_none=True)
        for p, gr in zip(params, grads):
            p.grad = gr

        def gstep():
            g.replay()
            if n > 1:
                hvd.grouped_allreduce_(grads, op=op, compression=compression,
                                       name="graph.grads")
            opt.step()
            return static_loss

        # One comm-free replay shakes out lazy pool plumbing before the
        # cross-rank mode agreement (no collective may be issued until
        # every rank has agreed to graph mode).
        g.replay()
        torch.cuda.synchronize()
        return gstep
    except Exception as e:  # pragma: no cover - GPU-only path
        print(f"[bench] hipGraph capture unavailable, eager fallback: {e}",
              flush=True)
        return None


def main():
    args = parse_args()
    hvd.init()
    n = hvd.size()
    rank = hvd.rank()
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda" if use_gpu else "cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32

    if args.model == "resnet50":
        model, opt, step, units_per_step, metric, cfg = build_resnet(args, device, dtype)
        default_comp = "none"
    elif args.model == "bert-large":
        model, opt, step, units_per_step, metric, cfg = build_bert(args, device, dtype)
        default_comp = "fp16"
    else:
        model, opt, step, units_per_step, metric, cfg = build_llama(
            args, device, dtype, args.model)
        default_comp = "none"

    comp_name = args.compression or default_comp
    compression = hvd.Compression.fp16 if comp_name == "fp16" \
        else hvd.Compression.none
    red = hvd.Adasum if args.reduction == "adasum" else hvd.Average
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters(),
                                   compression=compression, op=red)
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)

    for _ in range(args.warmup):
        step()

    # hipGraph replay for the launch-dense inner loop (see try_graph_step).
    # EXPERIMENTAL, opt-in via AXON_GRAPH=1: replay measures +16% on the
    # ResNet-50 headline config (8545 vs 7377 img/s same-process) and the
    # replayed gradients match eager bitwise at identical weights, but the
    # multi-step training trajectory still diverges from eager
    # (tools/graph_check.py fails), so results measured in this mode are
    # not yet honest training numbers and it must stay off by default.
    # Diagnosis so far (MI355X, ROCm 7.0 box): the delivery logic is
    # provably sound (a CPU emulation of stable-buffer grads + fused step
    # tracks eager exactly), and replay-at-initial-weights reproduces
    # eager grads bitwise — but after the first optimizer step the
    # replayed loss stops tracking live weights for part of the network
    # (trajectory consistent with convs frozen at their capture-time
    # values while the BN-affine/fc layers keep training). That pattern,
    # plus global-capture-mode hanging/segfaulting where thread_local
    # "works", indicates MIOpen conv setup work (weight transform /
    # workspace prep) escapes the captured stream and is snapshotted at
    # capture instead of replayed. Next lever: MIOpen find-mode pinning
    # or replacing the conv path before re-attempting default-on.
    graph_env = os.environ.get("AXON_GRAPH")
    want_graph = graph_env == "1"
    if want_graph and use_gpu and graph_env != "0":
        gstep = try_graph_step(opt, step.fwd, n, red, compression)
        ok = gstep is not None
        if n > 1:
            # All ranks must take the same mode: a mixed fleet issues
            # different collective name sets and stalls.
            flag = torch.tensor([1.0 if ok else 0.0], dtype=torch.float64)
            ok = hvd.allreduce(flag, op=hvd.Min,
                               name="bench.graph.ok").item() >= 1.0
        if ok:
            step = gstep
            if rank == 0:
                print("[bench] hipGraph replay mode on", flush=True)
        elif n > 1 and hasattr(opt, "_register_hooks") and \
                not getattr(opt, "_hook_handles", True):
            opt._register_hooks()  # capture path removed them; restore overlap

    if use_gpu:
        torch.cuda.synchronize()
    hvd.barrier(name="bench.start")
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    hvd.barrier(name="bench.stop")
    elapsed = time.perf_counter() - t0
    elapsed = hvd.allreduce(torch.tensor([elapsed], dtype=torch.float64),
                            op=hvd.Max, name="bench.elapsed").item()

    if rank == 0:
        value = units_per_step * n * args.steps / elapsed
        cfg["parallelism"] = f"dp{n}"
        cfg["reduction"] = args.reduction
        cfg["compression"] = comp_name
        print(json.dumps({
            "metric": metric,
            "value": round(value, 2),
            "unit": metric,
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "config": cfg,
        }), flush=True)
    hvd.shutdown()


if __name__ == "__main__":
    main()
