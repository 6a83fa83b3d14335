"""llm_instance_gateway_amd — an MI355X-native multi-tenant LLM inference gateway.

A from-scratch, single-node AMD Instinct MI355X (gfx950/CDNA4) re-design of the
capability surface of kubernetes-sigs/llm-instance-gateway (the early "LLM
Instance Gateway" / Gateway API Inference Extension):

  * the Envoy + ext-proc model-aware L7 router (reference:
    pkg/ext-proc/handlers/server.go, pkg/ext-proc/scheduling/scheduler.go)
    becomes an in-process router over the 8 GPUs of one node
    (`llm_instance_gateway_amd.router`);
  * the vLLM model-server fleet the reference delegates GPU work to becomes
    first-class per-GPU serving engines with paged-KV continuous batching and
    hand-written CDNA4 HIP kernels (`.engine`, `.models`, `.ops`);
  * InferencePool / InferenceModel CRD semantics (reference: api/v1alpha1/)
    become the file/API-driven registry (`.api`);
  * the dynamic-LoRA sidecar (reference: tools/dynamic-lora-sidecar/) becomes
    the in-process adapter lifecycle manager (`.lora`);
  * the SimPy cluster simulation (reference: simulations/llm_ig_simulation/)
    becomes a dependency-free discrete-event simulation (`.sim`);
  * tensor parallelism for large base models rides RCCL over xGMI
    (`.parallel`).

Import convention:  `import llm_instance_gateway_amd as lig`
"""

__version__ = "0.1.0"
