"""PipelineDriver / Operator lifecycle — backend-free CPU tests.

The driver restates PipelineDriver::process (reference
be/src/exec/runtime/pipeline_driver.cpp:391-500) and the Operator lifecycle
(exec_primitive/pipeline/operator.h:59-144); these tests pin the state
machine: pull/push pairing, finishing propagation down the chain, blocking
sinks that only produce at finish, and that no operator is pushed after it
stopped needing input.
"""

from starrocks_amd.pipeline import Operator, PipelineDriver


class ListSource(Operator):
    def __init__(self, items):
        super().__init__()
        self._items = list(items)

    def need_input(self):
        return False

    def has_output(self):
        return bool(self._items)

    def pull_chunk(self):
        return self._items.pop(0)


class Doubler(Operator):
    """One-in-one-out pass-through (a probe-shaped operator)."""

    def __init__(self, log):
        super().__init__()
        self._held = None
        self._log = log

    def need_input(self):
        return self._held is None and not self._finishing

    def has_output(self):
        return self._held is not None

    def push_chunk(self, chunk):
        assert not self._finishing, "push after finishing"
        self._held = chunk * 2
        self._log.append(("push", chunk))

    def pull_chunk(self):
        c, self._held = self._held, None
        self._log.append(("pull", c))
        return c


class CollectSink(Operator):
    def __init__(self):
        super().__init__()
        self.items = []

    def need_input(self):
        return not self._finishing

    def has_output(self):
        return False

    def push_chunk(self, chunk):
        self.items.append(chunk)

    def result(self):
        return self.items


def test_driver_moves_all_chunks_in_order():
    log = []
    sink = CollectSink()
    out = PipelineDriver([ListSource([1, 2, 3, 4]), Doubler(log), sink]).process()
    assert out == [2, 4, 6, 8]
    assert sink._closed and sink._finishing


def test_finishing_propagates_through_empty_source():
    out = PipelineDriver([ListSource([]), Doubler([]), CollectSink()]).process()
    assert out == []


def test_intermediate_drains_before_finishing():
    """An operator holding a chunk when upstream finishes must still drain it
    (is_finished requires !has_output, pipeline_driver.cpp's drain rule)."""
    log = []
    d = Doubler(log)
    out = PipelineDriver([ListSource([7]), d, CollectSink()]).process()
    assert out == [14]
    assert d.is_finished()


def test_long_chain():
    logs = [[] for _ in range(4)]
    ops = [ListSource(range(10))] + [Doubler(lg) for lg in logs] + [CollectSink()]
    out = PipelineDriver(ops).process()
    assert out == [i * 16 for i in range(10)]
