"""Mirror of reference test/test_library.py: the C-level self-test
(`bfTestSuite`, src/testsuite.cpp:189 in the reference) returns 0."""

from bifrost_amd.libbifrost_generated import bfTestSuite


def test_library():
    assert bfTestSuite() == 0
