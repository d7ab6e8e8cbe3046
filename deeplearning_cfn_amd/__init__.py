"""deeplearning_cfn_amd — MI355X-native distributed deep-learning framework.

A ground-up rebuild of the *capabilities* of ``awslabs/deeplearning-cfn``
(the AWS CloudFormation distributed-training launcher + its two worked
training recipes) for a single 8-GPU AMD Instinct MI355X node:

* the CloudFormation cluster provisioning + SQS worker handshake collapse
  into a topology preflight (:mod:`deeplearning_cfn_amd.topo`) and a local
  process-group rendezvous (:mod:`deeplearning_cfn_amd.launcher`);
* the documented env-var contract (``DEEPLEARNING_WORKERS_PATH``,
  ``DEEPLEARNING_WORKERS_COUNT``, ``DEEPLEARNING_WORKER_GPU_COUNT``,
  ``EFS_MOUNT``) is kept verbatim (:mod:`deeplearning_cfn_amd.contract`);
* the MXNet parameter-server and TensorFlow/Horovod data-parallel examples
  are re-realised as one PyTorch-ROCm DP engine whose gradient all-reduce
  runs on RCCL over xGMI with bucket fusion sized for the 7 point-to-point
  links per GPU (:mod:`deeplearning_cfn_amd.engine`);
* the hot ops are hand-written CDNA4 HIP kernels (MFMA GEMM/conv, fused
  BatchNorm+ReLU, fused SGD+momentum, NHWC<->NCHW pack) under
  :mod:`deeplearning_cfn_amd.ops`.

Reference provenance: the reference mount /root/reference was empty at
survey time; structural claims cite SURVEY.md sections,
which reconstruct the public repo (README.md sections,
cfn-template/deeplearning.template resources, examples/mxnet,
examples/tensorflow) with confidence tags.
"""

# NOTE: only the 3-channel stem conv still routes through MIOpen.  Its
# default (hybrid) find mode spends ~50 GPU-seconds benchmarking stem
# candidates once per process at batch 1024 — but MIOPEN_FIND_MODE=FAST
# was measured to select a naive kernel ~500x slower per step, so the
# one-time find cost is the right trade and no override is set here.


__version__ = "0.1.0"

from . import contract  # noqa: F401

__all__ = ["contract", "__version__"]
