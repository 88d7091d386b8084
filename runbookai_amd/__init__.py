"""runbookai_amd — MI355X-native AI SRE investigation framework.

A from-scratch rebuild of the capabilities of Runbook-Agent/RunbookAI
(reference: TypeScript CLI driving hosted LLM APIs) as an MI355X-first
stack: the agent runtime / knowledge base / tools / skills / eval layers
are Python, and every compute surface the reference bought from hosted
APIs (LLM chat+completion, text embeddings, vector search) runs locally
on AMD Instinct MI355X GPUs via PyTorch-ROCm plus hand-written CDNA4
(gfx950) HIP kernels, with RCCL over xGMI for tensor parallelism.

Reference layer map: /root/reference/src.
"""

__version__ = "0.1.0"
