import os

# Deterministic, fast MIOpen behaviour on fresh GPU boxes: immediate-mode
# find with the repo-shipped gfx950 DB (see bench.py).  Without this the
# suite can spend minutes in MIOpen kernel search for the small test convs.
_MIOPEN = os.path.join(os.path.dirname(os.path.abspath(__file__)), '..',
                       'chainermn_amd', '_miopen')
os.environ.setdefault('MIOPEN_FIND_MODE', 'FAST')
if os.path.isdir(_MIOPEN):
    os.environ.setdefault('MIOPEN_USER_DB_PATH',
                          os.path.join(_MIOPEN, 'db'))
    os.environ.setdefault('MIOPEN_CUSTOM_CACHE_DIR',
                          os.path.join(_MIOPEN, 'cache'))

import pytest  # noqa: F401,E402


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: needs a ROCm GPU (run on an MI355X box)')
