"""adversarial_spec_amd — MI355X-native adversarial spec debate engine.

A from-scratch rebuild of the capabilities of zscole/adversarial-spec
(reference layer map: SURVEY.md §1) where the remote-LLM fan-out
(reference: skills/adversarial-spec/scripts/models.py:681-722) is replaced
by on-node inference of open-weight opponent models on AMD Instinct MI355X
GPUs: hand-written CDNA4 (gfx950) HIP kernels for the hot ops, RCCL over
xGMI for the opponent-parallel consensus gather, PyTorch-ROCm for weight
hosting and plain library GEMMs (hipBLASLt).

Layers (MI355X mapping of SURVEY.md §1):
  cli/        - `debate.py` CLI surface (reference: debate.py:397-432)
  prompts     - prompt library (reference: prompts.py)
  providers   - provider/credential/registry/config tiers (reference: providers.py)
  session     - session + checkpoint store (reference: session.py)
  telegram    - human-in-the-loop channel (reference: telegram_bot.py)
  protocol    - [AGREE]/[SPEC]/[TASK] wire formats (reference: models.py:149-247)
  engine/     - round scheduler + opponent backends (stub / local GPU / litellm / CLI)
  models/     - Llama-family model definitions for the local engine
  ops/        - CDNA4 HIP kernels + CPU reference implementations
  parallel/   - RCCL-over-xGMI collectives: opponent consensus, tensor parallel
  utils/      - timing, synthetic spec generation
"""

__version__ = "1.2.0"
