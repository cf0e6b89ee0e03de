"""split_learning_amd — an MI355X-native split-learning training engine.

A from-scratch rebuild of the capabilities of filrg/split_learning
(see SURVEY.md for the structural analysis of the reference):

* models are partitioned at numbered cut layers (``start_layer < i <= end_layer``
  semantics, state-dict keys ``layerN.*`` — reference src/model/VGG16_CIFAR10.py:9-117)
  so saved ``{model}_{data}.pth`` files interoperate;
* cut-layer activations and gradients move between stages as GPU-resident
  tensors over RCCL point-to-point on xGMI side streams (replacing the
  reference's RabbitMQ/pickle transport, reference src/train/VGG16.py:20-53);
* per-partition math runs in hand-written CDNA4 (gfx950) HIP kernels —
  MFMA GEMMs and implicit-GEMM convolutions, fused BN/ReLU, fused loss and
  optimizers (the reference has no native code at all);
* per-round FedAvg aggregation is an RCCL all-reduce across same-stage GPUs
  (reference src/Utils.py:35-66, src/Server.py:398-434);
* the ``config.yaml`` schema, ``server.py``/``client.py`` entrypoints, and the
  round protocol (REGISTER/START/SYN/NOTIFY/PAUSE/UPDATE/STOP,
  reference src/Server.py:103-212) stay compatible.
"""

__version__ = "0.1.0"
