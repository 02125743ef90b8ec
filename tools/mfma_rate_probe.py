"""fp8-vs-bf16 MFMA rate on THIS chip (config-5 attention decision)."""
import sys
import time

sys.path.insert(0, "/root/repo")
import torch  # noqa: E402

from adversarial_spec_amd.ops import _load_hip  # noqa: E402

hip = _load_hip()
seed = torch.randn(2048, device="cuda").bfloat16()
out = torch.zeros(4096, device="cuda", dtype=torch.float32)
BLOCKS = 2048  # 8 blocks/CU
ITERS, ACCS, WAVES = 4096, 8, BLOCKS * 4
FLOPS = 2 * 16 * 16 * 32 * ITERS * ACCS * WAVES


def run(which):
    for _ in range(2):
        hip.mfma_rate(seed, out, which, BLOCKS)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        hip.mfma_rate(seed, out, which, BLOCKS)
    torch.cuda.synchronize()
    return FLOPS * 5 / (time.perf_counter() - t0)


for _ in range(3):  # interleaved rounds (guide rule 24)
    tb = run(16)
    tf = run(8)
    print(f"bf16 16x16x32: {tb/1e12:7.1f} TF/s | fp8 16x16x32: "
          f"{tf/1e12:7.1f} TF/s | fp8/bf16 = {tf/tb:.3f}", flush=True)
