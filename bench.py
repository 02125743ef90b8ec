#!/usr/bin/env python3
"""Flagship benchmark: debate-round wall-clock + critiques/sec on MI355X.

Measures BASELINE.json's headline metric — one adversarial debate round of
N_opp opponents critiquing an 8k-token synthetic spec — on random-init
Llama-3-8B opponents (bf16), exactly the configuration BASELINE.json names
(config 3 at world_size 1: "3 opponents, 8k-token spec").

Scaling is WEAK: every GPU hosts `--opponents-per-gpu` (default 3)
co-resident opponents in its 288 GB of HBM3E; adding GPUs adds opponents.
A round = every opponent prefills the spec prompt and decodes its critique,
then one fused RCCL all-gather over xGMI collects all critique token
buffers + agreed flags for the consensus check (parallel/consensus.py).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from adversarial_spec_amd.engine.local import LocalEngine  # noqa: E402
from adversarial_spec_amd.engine.scheduler import build_user_message  # noqa: E402
from adversarial_spec_amd.parallel.consensus import (  # noqa: E402
    HDR,
    AsyncRoundGather,
    pack_result,
    unpack_results,
)
from adversarial_spec_amd.prompts import get_system_prompt  # noqa: E402
from adversarial_spec_amd.protocol import detect_agreement  # noqa: E402
from adversarial_spec_amd.utils.synth import synthetic_spec  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3, help="timed debate rounds")
    p.add_argument("--warmup", type=int, default=1, help="untimed rounds")
    p.add_argument("--opponents-per-gpu", type=int, default=3)
    p.add_argument("--spec-tokens", type=int, default=8192)
    p.add_argument("--decode-tokens", type=int, default=512,
                   help="critique length decoded per opponent per round "
                        "(the reference caps critiques at 8000 tokens, "
                        "models.py:620; 512 is a representative critique+"
                        "[SPEC] revision length for the headline round)")
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--arch-mix", default=None,
                   help="comma list of archs cycled over opponents for the "
                        "heterogeneous config (e.g. llama-3-8b,mistral-7b)")
    p.add_argument("--temperature", type=float, default=0.7)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp8"],
                   help="projection compute dtype (fp8 = BASELINE config 5)")
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree: all ranks form ONE sharded "
                        "opponent (BASELINE config 5: llama-3-70b --tp 8)")
    p.add_argument("--cpu-smoke", action="store_true",
                   help="validation mode: tiny opponents on CPU over gloo — "
                        "exercises the full multi-rank round flow (threaded "
                        "opponents, ordered consensus gathers) without GPUs")
    return p.parse_args()


def main() -> int:
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    use_gpu = torch.cuda.is_available() and not args.cpu_smoke
    if not use_gpu and not args.cpu_smoke:
        print("bench.py requires an MI355X GPU (or --cpu-smoke)", file=sys.stderr)
        return 1
    if use_gpu:
        # ranks may outnumber GPUs (e.g. torchrun 2 ranks on a 1-GPU box to
        # exercise RCCL over a real communicator): wrap into the device pool
        dev_idx = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev_idx)
        device = f"cuda:{dev_idx}"
        # decode-attention split-block target is picked by the ENGINE per
        # dtype regime (LocalEngine._pick_split_blocks, A/B-measured);
        # ADVSPEC_SPLIT_BLOCKS env remains the explicit override.
    else:
        device = "cpu"
        args.model = "tiny"
        args.arch_mix = None
        args.spec_tokens = min(args.spec_tokens, 256)
        args.decode_tokens = min(args.decode_tokens, 8)

    # Always run the consensus gather through a REAL communicator — at
    # world 1 too: a single-rank nccl group still executes the fused
    # all-gather as an RCCL collective on-device, so the measured round
    # includes the collective cost at every N and the metric label is
    # honest (round-1 verdict: the N=1 path silently skipped the gather).
    import torch.distributed as dist_mod

    dist = dist_mod
    if world == 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29371")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
    dist.init_process_group("nccl" if use_gpu else "gloo")

    tp_mode = args.tp > 1
    if tp_mode:
        if dist is None or world != args.tp:
            print("--tp N needs a torchrun world of exactly N ranks", file=sys.stderr)
            return 1
        from adversarial_spec_amd.parallel.tp import TPContext

        n_opp = 1
        engines = [
            LocalEngine(
                {"name": f"{args.model}-tp", "arch": args.model,
                 "dtype": args.dtype},
                device=device, tp=TPContext.from_default_group(),
            )
        ]
    else:
        n_opp = args.opponents_per_gpu
        archs = (args.arch_mix.split(",") if args.arch_mix else [args.model])
        engines = []
        for i in range(n_opp):
            # distinct seeds per (rank, opponent): heterogeneous random-init pool
            arch = archs[(rank * n_opp + i) % len(archs)]
            engines.append(
                LocalEngine(
                    {"name": f"{arch}-r{rank}o{i}", "arch": arch,
                     "dtype": args.dtype},
                    device=device,
                )
            )

    spec = synthetic_spec(args.spec_tokens, seed=17)
    system_prompt = get_system_prompt("tech")
    user_message = build_user_message(spec, 1, "tech")
    max_gather_tokens = args.decode_tokens

    # One fused all-gather per OPPONENT, launched on a dedicated comm
    # stream as soon as that opponent's decode finishes: opponent i's
    # consensus traffic overlaps opponent i+1's prefill/decode (the
    # BASELINE north-star overlap; collectives are issued in the same
    # opponent order on every rank).
    gathers = (
        [AsyncRoundGather(max_tokens=max_gather_tokens) for _ in range(n_opp)]
        if (dist is not None and not tp_mode) else None
    )

    pool = ThreadPoolExecutor(max_workers=max(1, n_opp))

    def run_one(eng):
        # per-opponent fault isolation (reference semantics: an errored
        # opponent degrades the round, never aborts it — and on the
        # distributed path an un-launched collective would DEADLOCK the
        # other ranks, so errors must still reach the gather)
        try:
            text, _in, _out, _tm = eng.generate(
                system_prompt,
                user_message,
                max_tokens=args.decode_tokens,
                temperature=args.temperature,
                timeout=600.0,
            )
            ids = eng.tokenizer.encode(text)[: max_gather_tokens]
            agreed = detect_agreement(text)
            return ids, agreed, False
        except Exception as e:  # noqa: BLE001
            print(f"opponent {eng.name} failed: {e}", file=sys.stderr)
            return [], False, True

    def one_round():
        """One debate round for this rank's opponents + consensus gather.

        Co-resident opponents run CONCURRENTLY: each engine issues its
        prefill/decode on its own HIP stream from its own host thread, so
        the decode GEMVs of N opponents interleave and fill HBM bandwidth
        that a single bandwidth-bound decode leaves idle. Collectives are
        still launched in opponent order (identical on every rank).
        """
        local = []
        futures = [pool.submit(run_one, eng) for eng in engines]
        for i, fut in enumerate(futures):
            ids, agreed, error = fut.result()
            if gathers is not None:
                gathers[i].launch(ids, agreed, error)
            else:
                local.append(
                    pack_result(ids, agreed, error, max_gather_tokens,
                                torch.device(device))
                )
        if gathers is not None:
            results = []
            all_agreed = True
            for g in gathers:
                r, ok = g.wait()
                results.extend(r)
                all_agreed = all_agreed and ok
            return all_agreed
        rows = torch.cat(local).view(n_opp, HDR + max_gather_tokens)
        results = unpack_results(rows)
        ok = [r for r in results if not r.error]
        return bool(ok) and all(r.agreed for r in ok)

    # warmup
    for _ in range(args.warmup):
        one_round()

    if dist is not None:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_round()
    if dist is not None:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist is not None:
        e = torch.tensor([elapsed], device=device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    n_job_opponents = n_opp if tp_mode else world * n_opp
    total_critiques = n_job_opponents * args.steps
    value = total_critiques / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    consensus = (
        "local consensus (TP ranks form one opponent)" if tp_mode
        else ("RCCL consensus all-gather" if use_gpu
              else "gloo consensus all-gather")
    )
    if rank == 0:
        print(json.dumps({
            "metric": f"critiques/sec (debate round: {args.spec_tokens}-token "
                      f"spec prefill + {args.decode_tokens}-token critique "
                      f"decode + {consensus})",
            "value": value,
            "unit": "critiques/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": args.model,
                "opponents": n_job_opponents,
                "opponents_per_gpu": n_opp if not tp_mode else 1.0 / world,
                "spec_tokens": args.spec_tokens,
                "decode_tokens": args.decode_tokens,
                "global_batch": n_job_opponents,
                "seq_len": args.spec_tokens,
                "parallelism": (f"tp{args.tp}" if tp_mode
                                else f"opponent-parallel dp{world}"),
                "temperature": args.temperature,
            },
        }))

    if dist is not None:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
