"""`python -m bifrost_amd.version [--config]` — version / build
configuration banner (reference python/bifrost/version/__main__.py
surface, reported for the MI355X/HIP backend)."""

import argparse
import os

from bifrost_amd.version import __version__
from bifrost_amd import libbifrost_generated as cfg


def _yes_no(value):
    return "yes" if value else "no"


def main():
    parser = argparse.ArgumentParser(
        description="Bifrost (MI355X backend) version/configuration "
                    "information")
    parser.add_argument("--config", action="store_true",
                        help="also display configuration information")
    args = parser.parse_args()

    print("bifrost_amd " + __version__)
    print("MI355X-native (gfx950 / CDNA4) backend for the bifrost C ABI")
    if args.config:
        lib = os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "lib", "libbifrost.so")
        print("\nConfiguration:")
        print(" Library: %s" % lib)
        print(" GPU support (HIP/ROCm): %s" % _yes_no(cfg.BF_CUDA_ENABLED))
        print(" GPU architectures: gfx950")
        print(" Debugging: %s" % _yes_no(cfg.BF_DEBUG_ENABLED))
        print(" Tracing: %s" % _yes_no(cfg.BF_TRACE_ENABLED))
        print(" float128 support: %s" % _yes_no(cfg.BF_FLOAT128_ENABLED))
        print(" Map kernel disk cache: %s" %
              _yes_no(os.environ.get("BIFROST_NO_DISK_CACHE", "0") != "1"))
        print(" Logging directory: %s" %
              os.environ.get("BIFROST_PROCLOG_DIR",
                             "/dev/shm/bifrost_amd"))


if __name__ == "__main__":
    main()
