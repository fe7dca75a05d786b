"""Driver loop — the host analog of Driver.processInternal
(presto-main-base/.../operator/Driver.java:402-475): moves pages through a
chain of operators with the needsInput/addInput/getOutput protocol, one
page at a time, finishing upstream operators before draining downstream.

The GPU operators buffer/batch internally (the Operator contract permits
it), so this loop is the fidelity surface: it never calls addInput unless
needsInput is true (Driver.java:446-458), drains getOutput after every
push, and propagates finish() down the chain.
"""
from .engine import Operator, Page, PgPage


def run_chain(source_pages, ops, sink):
    """Drive `source_pages` (iterable of Page) through `ops` (list of
    Operator); call `sink(raw_page)` for every output page of the last
    operator.  Mirrors the two-operator pump of Driver.processInternal:
    for each adjacent pair, move current.getOutput() into next.addInput().
    """
    n = len(ops)

    def drain(from_idx):
        """move any pending outputs downstream from ops[from_idx]"""
        for i in range(from_idx, n):
            while True:
                out = ops[i].get_output_raw()
                if out is None:
                    break
                if i + 1 < n:
                    assert ops[i + 1].needs_input()
                    ops[i + 1].add_input_raw(out)
                else:
                    sink(out)

    for page in source_pages:
        assert ops[0].needs_input()
        ops[0].add_input(page)
        drain(0)
    for i in range(n):
        ops[i].finish()
        drain(i)
    for op in ops:
        assert op.is_finished()
