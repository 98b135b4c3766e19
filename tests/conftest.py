import os
import sys

# make the in-tree package importable regardless of the invoking cwd
_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
for p in (_REPO, os.path.dirname(os.path.abspath(__file__))):
    if p not in sys.path:
        sys.path.insert(0, p)


def pytest_configure(config):
    config.addinivalue_line(
        'markers', 'gpu: requires an MI355X GPU (run on a gpurun box)')
