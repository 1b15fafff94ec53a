"""crowdllama-amd: MI355X-native peer-to-peer LLM inference mesh.

A from-scratch reimplementation of the capabilities of crowdllama/crowdllama
(reference: a Go libp2p mesh delegating compute to Ollama), redesigned for
AMD Instinct MI355X (gfx950 / CDNA4):

- mesh/        — DHT-style rendezvous, peer discovery, peer manager, gateway
                 (Ollama-compatible /api/chat), protobuf wire protocol, IPC.
                 (reference layers L2-L7: pkg/peer, pkg/dht, pkg/gateway,
                 pkg/peermanager, internal/discovery, pkg/ipc)
- engine/      — the worker inference engine replacing the reference's
                 shell-out to Ollama (reference L1: pkg/crowdllama/api.go),
                 backed by hand-written HIP/CDNA4 kernels.
- ops/         — HIP kernels + C++ runtime (GGUF loader, decode engine,
                 hipGraph capture) for gfx950.
- quant/       — GGUF v3 reader/writer and K-quant (Q4_K/Q6_K/Q8_0) CPU
                 reference codecs.
- models/      — model-family presets (llama3, mistral, tinyllama) and
                 synthetic random-init checkpoint generation.
- parallel/    — tensor-parallel sharding over RCCL/xGMI.
"""

from .version import __version__  # noqa: F401
