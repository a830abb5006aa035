"""HIP/CDNA4 kernel extension helpers (device learner build metadata lives here)."""
