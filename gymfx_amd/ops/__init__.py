"""HIP/CDNA4 native ops for gymfx_amd (gfx950-only, no CUDA shims)."""
