"""MXNet plugin — not provided; see byteps_amd.tensorflow.__init__ for
the rationale (PyTorch-ROCm-first build; MXNet not in the target image)."""

raise ImportError(
    "byteps_amd targets PyTorch-ROCm (use byteps_amd.torch); "
    "no MXNet runtime exists in this environment.")
