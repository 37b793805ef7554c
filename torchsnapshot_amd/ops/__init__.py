"""Device data-plane ops: HIP/CDNA4 staging engine for gfx950."""
