"""Build the in-tree gfx950 HIP extension: python -m code_intelligence_amd.ops.build"""
from . import extension

if __name__ == "__main__":
    path = extension.build(verbose=True)
    print(f"built {path}")
