"""PrintHeaderBlock (reference blocks/print_header.py surface): print
each new sequence's header — a quick debugging sink."""

import pprint
from threading import Lock

from bifrost_amd.pipeline import SinkBlock

__all__ = ["PrintHeaderBlock", "print_header"]


class PrintHeaderBlock(SinkBlock):
    lock = Lock()

    def on_sequence(self, iseq):
        with PrintHeaderBlock.lock:
            print("-----")
            print("Block", self.iring.owner.name, iseq.header.get("name"))
            pprint.pprint(iseq.header)
            print("-----")

    def on_data(self, ispan):
        pass


def print_header(iring, *args, **kwargs):
    """Print the header of each new sequence (testing/debugging aid)."""
    return PrintHeaderBlock(iring, *args, **kwargs)
