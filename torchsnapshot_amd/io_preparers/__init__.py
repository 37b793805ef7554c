"""Type-specific I/O preparers: runtime object <-> manifest entry + I/O reqs."""
