"""Device op helpers (gfx950 kernels behind torch-tensor interfaces)."""
from splatt_amd.ops.dense import gram, spd_inverse

__all__ = ["gram", "spd_inverse"]
