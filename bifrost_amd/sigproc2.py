"""Alias module for the reference's `bifrost.sigproc2` import path: the
v1/v2 sigproc surfaces are merged in `bifrost_amd.sigproc`."""

from bifrost_amd.sigproc import *  # noqa: F401,F403
from bifrost_amd.sigproc import (read_header, write_header,  # noqa: F401
                                 seek_to_data, pack, unpack,
                                 id2telescope, telescope2id, id2machine,
                                 machine2id, SigprocFile)
