import io
import os
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if REPO_ROOT not in sys.path:
    sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: requires an MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def fixture_tree(tmp_path_factory):
    """The standard deterministic test fixture tree."""
    from dragnet_amd.tools.mktestdata import make_fixture_tree
    root = tmp_path_factory.mktemp("data")
    make_fixture_tree(str(root))
    return str(root)


class CliResult(object):
    def __init__(self, code, out, err):
        self.code = code
        self.out = out
        self.err = err


@pytest.fixture
def dn(tmp_path, monkeypatch):
    """Run the dn CLI in-process with an isolated config file.

    Usage: result = dn('scan', '-b', 'operation', 'src')
    """
    cfgfile = str(tmp_path / "dragnet_config.json")
    monkeypatch.setenv("DRAGNET_CONFIG", cfgfile)
    monkeypatch.setenv("DRAGNET_ENGINE",
                       os.environ.get("DRAGNET_ENGINE", "cpu"))

    def run(*argv, stdin=b""):
        from dragnet_amd import cli
        old_out, old_err = sys.stdout, sys.stderr
        old_in = sys.stdin
        out, err = io.StringIO(), io.StringIO()
        sys.stdout, sys.stderr = out, err
        sin = io.TextIOWrapper(io.BytesIO(stdin))
        sys.stdin = sin
        try:
            code = cli.main(list(argv))
        finally:
            sys.stdout, sys.stderr = old_out, old_err
            sys.stdin = old_in
        return CliResult(code, out.getvalue(), err.getvalue())

    return run
