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

# NOTE (round 2): the 3-channel stem conv now runs natively via the
# small-Cin packed path (ops/csrc/conv.hip StemGather) —
# fwd, fused conv+BN stats, and wrw.  MIOpen is reached only under
# DLX_CONV=torch; its hybrid find mode costs ~50 GPU-seconds once per
# process there (MIOPEN_FIND_MODE=FAST picks a ~500x slower kernel, so
# no override is set).


__version__ = "0.1.0"

from . import contract  # noqa: F401

__all__ = ["contract", "__version__"]
