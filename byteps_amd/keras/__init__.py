"""Keras plugin — not provided; see byteps_amd.tensorflow.__init__ for
the rationale."""

raise ImportError(
    "byteps_amd targets PyTorch-ROCm (use byteps_amd.torch); "
    "no TensorFlow/Keras runtime exists in this environment.")
