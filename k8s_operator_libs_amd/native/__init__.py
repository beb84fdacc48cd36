"""Native HIP/gfx950 components (built in-tree via build.py)."""
