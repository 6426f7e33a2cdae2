"""Version info (reference bifrost.version surface, hand-maintained —
there is no configure step here)."""

__version__ = "0.1.0"
__branch__ = "mi355x-native"

CONFIG = {
    "backend": "HIP/ROCm (gfx950)",
    "cuda_enabled": True,   # device support via HIP; 'cuda' space alias
    "float128_enabled": False,
    "debug_enabled": False,
}
