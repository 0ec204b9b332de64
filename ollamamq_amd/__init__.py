"""ollamamq_amd — MI355X-native multi-user LLM dispatcher + inference engine.

A from-scratch rebuild of the capabilities of Chleba/ollamaMQ (a Rust HTTP
message-queue dispatcher in front of external Ollama/LM Studio backends,
see /root/reference) re-designed MI355X-first:

* the dispatcher (queues / fair-share scheduler / health / control plane /
  admin API / TUI) is native C++ (``csrc/dispatcher``), wire-compatible with
  the reference's Ollama + OpenAI API surface (reference src/main.rs:264-308);
* each "backend" is an in-process GPU worker running a hand-written CDNA4
  HIP inference engine (prefill/decode attention, RMSNorm, RoPE, SwiGLU,
  sampler) on one MI355X, instead of an external HTTP inference server;
* large models span GPUs as one logical backend via tensor-parallel RCCL
  all-reduce over xGMI (one process per GPU, torch.distributed "nccl").
"""

__version__ = "0.1.0"
