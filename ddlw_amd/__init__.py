"""ddlw_amd — MI355X-native distributed image-classification framework.

A from-scratch rebuild of the capabilities of
smellslikeml/distributed-deep-learning-workshop (Spark + Petastorm + TF/Keras +
Horovod + Hyperopt + MLflow on Databricks) as a single-node, 8×MI355X-native
stack:

- ``ddlw_amd.core``     — config dataclasses + MLflow-layout file tracking
                          store / model registry (reference: MLflow usage in
                          ``Part 2 .../01_hyperopt_single_machine_model.py:221-299``).
- ``ddlw_amd.data``     — JPEG tree -> bronze/silver Parquet pipeline and the
                          row-group-sharded streaming loader (reference:
                          ``Part 1 .../01_data_prep.py``, Petastorm usage in
                          ``Part 1 .../03_model_training_distributed.py:135-144``).
- ``ddlw_amd.models``   — ``build_model()`` equivalents: SmallCNN, ResNet-50,
                          MobileNetV2-head (reference:
                          ``Part 1 .../02_model_training_single_node.py:159-178``).
- ``ddlw_amd.ops``      — hand-written CDNA4 HIP kernels (MFMA implicit-GEMM
                          conv, fused BN+ReLU, maxpool, softmax-CE, fused SGD)
                          behind ``torch.autograd.Function``.
- ``ddlw_amd.parallel`` — RCCL-over-xGMI data parallelism: ``hvd``-style API
                          (init/rank/size/DistributedOptimizer/broadcast) and a
                          local multi-process launcher (reference: Horovod call
                          sites in ``Part 1 .../03_model_training_distributed.py:282-375``).
- ``ddlw_amd.train``    — Keras-like ``Model.compile/fit/evaluate`` facade +
                          callbacks (reference: ``.../02_model_training_single_node.py:198-215``).
- ``ddlw_amd.tune``     — TPE ``fmin`` + ``hp`` search-space DSL + LocalTrials
                          (reference: ``Part 2 .../01_hyperopt_single_machine_model.py:194-238``).
- ``ddlw_amd.infer``    — packaged predict-function (pyfunc equivalent) +
                          multi-GPU predict-UDF fan-out (reference:
                          ``Part 2 .../03_pyfunc_distributed_inference.py:157-234,466-472``).

Design target: MI355X (gfx950, CDNA4) only — PyTorch-ROCm for the module
system/autograd, hand-written HIP for the hot ops, RCCL over the 7-link xGMI
mesh for collectives. No CUDA path, no Triton, no hipify.
"""

__version__ = "0.1.0"
