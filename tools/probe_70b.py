"""BASELINE config-5 single-GPU probe: Llama-3-70B fp8 on one MI355X.

The full config (TP=8, 32k prefill) needs the 8-GPU node the driver may
or may not have; this probe measures what ONE GPU supports: the 70B fp8
opponent resident in 288 GB HBM3E, timed prefill at the requested length
(default 4096 — a 32k x 70B prefill is ~1 GPU-hour of math on one chip,
i.e. exactly why config 5 is TP=8) and per-token fp8 decode. Prints one
JSON line per phase for profiles/.
"""
import json
import sys
import time

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from adversarial_spec_amd.engine.local import LocalEngine  # noqa: E402


def main():
    prefill_len = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
    decode_toks = int(sys.argv[2]) if len(sys.argv) > 2 else 24
    t0 = time.perf_counter()
    eng = LocalEngine({"name": "probe-70b", "arch": "llama-3-70b",
                       "dtype": "fp8"}, device="cuda:0")
    torch.cuda.synchronize()
    init_s = time.perf_counter() - t0
    mem = torch.cuda.memory_allocated() / 2**30
    print(json.dumps({"phase": "init", "model": "llama-3-70b", "dtype": "fp8",
                      "seconds": round(init_s, 1),
                      "hbm_gib": round(mem, 1)}), flush=True)

    cache = eng._get_cache(prefill_len + decode_toks + 16)
    prompt = torch.randint(0, 255, (prefill_len,), device="cuda",
                           dtype=torch.long)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    logits = eng.model.prefill(prompt, cache)
    torch.cuda.synchronize()
    pre_s = time.perf_counter() - t0
    print(json.dumps({"phase": "prefill", "tokens": prefill_len,
                      "seconds": round(pre_s, 2),
                      "tok_per_s": round(prefill_len / pre_s, 1)}),
          flush=True)

    tok = int(logits.argmax().item())
    # warm decode
    for _ in range(4):
        logits = eng.model.decode_one(tok, cache)
        tok = int(logits.argmax().item())
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(decode_toks):
        logits = eng.model.decode_one(tok, cache)
        tok = int(logits.argmax().item())
    torch.cuda.synchronize()
    dec_s = time.perf_counter() - t0
    print(json.dumps({"phase": "decode", "tokens": decode_toks,
                      "ms_per_tok": round(dec_s / decode_toks * 1e3, 2),
                      "tok_per_s": round(decode_toks / dec_s, 2)}),
          flush=True)

    # the PRODUCTION decode path: engine generate() = workspace step +
    # quant_norm fusion + HIP-graph replay (decode_one above is the eager
    # per-token loop with per-call allocations — a correctness anchor,
    # not the serving path)
    spec_words = "api endpoint latency budget " * (prefill_len // 5)
    gen_toks = max(64, decode_toks)
    for label in ("generate(cold: graph capture)", "generate(warm graph)"):
        t0 = time.perf_counter()
        _text, _i, out_n, tm = eng.generate(
            "You are a reviewer.", spec_words, max_tokens=gen_toks,
            temperature=0.0, timeout=1800.0,
        )
        tot = time.perf_counter() - t0
        dec_ms = tm.get("decode", 0.0)  # PhaseTimer reports milliseconds
        print(json.dumps({"phase": label, "out_tokens": out_n,
                          "ms_per_tok": round(dec_ms / max(out_n, 1), 2),
                          "total_s": round(tot, 2)}), flush=True)


if __name__ == "__main__":
    main()
