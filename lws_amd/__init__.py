"""lws_amd — MI355X-native LeaderWorkerSet/DisaggregatedSet framework.

A standalone, from-scratch re-design of kubernetes-sigs/lws for AMD
Instinct MI355X deployments: the same LeaderWorkerSet v1 and
DisaggregatedSet v1 API surface and orchestration semantics, running on an
in-repo control plane, driving an in-repo PyTorch-ROCm + HIP/CDNA4 + RCCL
serving engine.
"""
__version__ = "0.1.0"
