"""baton-mi355x: an MI355X-native federated-learning engine.

A from-scratch re-design of the capabilities of mynameisfiber/baton
(/root/reference — an aiohttp + PyTorch FedAvg coordinator): an HTTP control
plane (registration, heartbeat/TTL liveness, synchronous round orchestration,
same routes / payload schemas / status codes as the reference, see SURVEY.md
§2.4) plus an MI355X data plane: the 8 GPUs of one node act as 8 federated
clients, each local training loop runs on its own GPU with the hot ops
hand-written as CDNA4 HIP (gfx950) kernels, and the FedAvg weighted mean +
global-model broadcast run as RCCL reduce + broadcast over xGMI.

Layering (mirrors SURVEY.md §1, re-designed):
  utils/    L0 — async scaffolding, keys, config, tracing
  control/  L1+L2 — HTTP control plane: registry, round state machine,
            manager (Experiment), worker runtime, wire format
  fed/      aggregation math (FedAvg / FedProx), data splitters
  ops/      L(-1) — HIP/CDNA4 kernels + torch.autograd wrappers
  models/   L3 — model zoo (MLP demo, ResNet, BERT, Llama-LoRA)
  parallel/ RCCL-over-xGMI data plane (one process per GPU)
  runtime/  local trainer, hipGraph-captured step
"""

__version__ = "0.1.0"

from baton_amd.utils.config import BatonConfig  # noqa: F401
