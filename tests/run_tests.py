"""Test runner (parity: the reference's tests/run_tests.py)."""
import sys

import pytest

if __name__ == "__main__":
    sys.exit(pytest.main(["-q", "-m", "not gpu", "tests", *sys.argv[1:]]))
