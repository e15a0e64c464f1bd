"""Driver entry-point contracts (__graft_entry__.py)."""

import pytest


def test_smoke_subprocess_plumbing_cpu():
    """smoke() runs the GPU step in a subprocess and must propagate a
    REAL child failure (here: no GPU in this container) as an exception
    carrying the child's error — not swallow it as a teardown crash."""
    import __graft_entry__ as g

    with pytest.raises(RuntimeError) as e:
        g.smoke()
    msg = str(e.value)
    assert "smoke step failed" in msg
    assert "needs a GPU" in msg  # the child's actual assertion surfaced
