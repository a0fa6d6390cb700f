"""gpustack_amd — MI355X-native model-serving cluster manager.

A from-scratch framework with the capabilities of GPUStack:
FastAPI control plane (server, scheduler, OpenAI-compatible gateway, auth,
observability) + GPU workers running a first-party CDNA4/HIP inference
engine (paged attention, RMSNorm, RoPE, sampling on MFMA; RCCL over xGMI
for tensor parallelism; 288 GB HBM3E-sized KV pools).
"""
__version__ = "0.1.0"
