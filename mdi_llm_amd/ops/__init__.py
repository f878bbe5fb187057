"""HIP/CDNA4 kernel extension loader (populated as kernels land)."""
