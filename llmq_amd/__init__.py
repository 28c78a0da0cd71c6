"""llmq-amd — MI355X-native distributed batch-inference framework.

A from-scratch rebuild of the capabilities of iPieter/llmq (reference:
/root/reference) designed MI355X-first:

- An in-tree asyncio message broker (``llmq_amd.broker``) replaces the
  external RabbitMQ dependency (reference: llmq/core/broker.py talks AMQP to
  an external rabbitmq process). Durable job/result queues, per-consumer
  prefetch, ack/requeue, and a real dead-letter queue with a retry cap
  (the reference's ``.failed`` queue is read-only scaffolding).
- An in-tree inference engine (``llmq_amd.engine``) replaces the vLLM
  dependency (reference: llmq/workers/vllm_worker.py). Continuous batching,
  paged KV cache sized for 288 GB HBM3E, hand-written CDNA4 HIP kernels
  (RMSNorm, RoPE, paged attention prefill/decode, SiLU·mul, sampling) on
  MFMA with LDS staging, hipGraph-captured decode, RCCL-over-xGMI tensor
  parallelism.
- The same CLI surface: submit / receive / status / health / errors /
  clear / worker {run,dummy,semhash,pipeline} and YAML multi-stage
  pipelines.
"""

__version__ = "0.1.0"
