"""TensorFlow plugin — not provided in the MI355X-native build.

The reference shipped TensorFlow/Keras/MXNet plugins (reference
byteps/tensorflow, byteps/keras, byteps/mxnet) because its engine lived
below the framework layer.  This rebuild is PyTorch-ROCm-first per its
north star (BASELINE.json): the engine's framework-facing contract is
``byteps_amd.torch``.  The layers a TF-ROCm plugin would need — name/key
registry, partitioning, priority engine, KV client, codecs — are all
framework-agnostic (byteps_amd.common / byteps_amd.ops); this module
documents the mapping and fails loudly rather than shipping untestable
code (TensorFlow is not installed in the target image).
"""

raise ImportError(
    "byteps_amd targets PyTorch-ROCm (use byteps_amd.torch). "
    "A TensorFlow-ROCm plugin would bind tf.ops to the same engine "
    "(byteps_amd.common + byteps_amd.ops); TensorFlow is not available "
    "in this environment, so no untested binding is shipped.")
