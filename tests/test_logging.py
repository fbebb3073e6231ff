"""Debug-log format tests.

The reference asserts its `r<rank> | <id> | <Op>` format via capsys
(``tests/test_common.py:118-146``); we keep the same format and toggle.
"""

import re

import torch

import mpi4jax_amd as m


def test_log_format(capfd):
    m.set_logging(True)
    try:
        m.allreduce(torch.zeros(3), m.SUM)
    finally:
        m.set_logging(False)
    out, _ = capfd.readouterr()
    lines = [l for l in out.splitlines() if l.startswith("r0")]
    assert len(lines) >= 2
    assert re.match(r"^r0 \| [a-z0-9]{8} \| Allreduce \(3 items\)$", lines[0])
    assert re.match(
        r"^r0 \| [a-z0-9]{8} \| done with code 0 \(\d\.\d{2}e[+-]\d{2}s\)$",
        lines[1],
    )


def test_logging_toggle(capfd):
    m.set_logging(False)
    m.allreduce(torch.zeros(3), m.SUM)
    out, _ = capfd.readouterr()
    assert "Allreduce" not in out
    assert m.get_logging() is False
    m.set_logging(True)
    assert m.get_logging() is True
    m.set_logging(False)


def test_env_toggle(monkeypatch):
    # logging module reads MPI4JAX_AMD_DEBUG at import
    import importlib

    import mpi4jax_amd.utils.logging as lg

    monkeypatch.setenv("MPI4JAX_AMD_DEBUG", "1")
    importlib.reload(lg)
    assert lg.get_logging() is True
    monkeypatch.delenv("MPI4JAX_AMD_DEBUG")
    importlib.reload(lg)
    assert lg.get_logging() is False
