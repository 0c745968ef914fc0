"""predictionio_amd — MI355X-native prediction-engine server.

A from-scratch re-design of the capabilities of apache/incubator-predictionio
(reference: /root/reference, Scala/Spark Lambda-architecture ML server) as an
MI355X-first framework:

- Event layer + pluggable storage (reference: data/.../storage/*.scala)
- DASE controller API: DataSource / Preparator / Algorithm / Serving
  (reference: core/.../controller/*.scala)
- Workflow runtime: train / deploy / batch-predict / eval
  (reference: core/.../workflow/*.scala)
- Compute core: hand-written gfx950 HIP kernels (ALS Gramian+Cholesky,
  masked top-K scoring GEMM, cosine kNN) replacing Spark MLlib, dispatched
  through PyTorch-ROCm tensors; multi-GPU data parallelism over RCCL/xGMI
  replacing Spark shuffles.

The reference's compute substrate (Spark executors + MLlib + netlib BLAS) is
replaced wholesale; its API surface (event REST schema, engine.json, DASE
semantics, CLI verbs, model checkpoint behavior) is reproduced.
"""

__version__ = "0.1.0"
